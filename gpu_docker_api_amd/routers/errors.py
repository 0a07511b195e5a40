"""Exception -> business-code mapping (the reference does this per-handler
with errors.Cause switches, e.g. routers/replicaset.go:154-169)."""
from __future__ import annotations

import logging

from .. import xerrors
from ..models.memory import UnsupportedSizeUnit
from .codes import Code

log = logging.getLogger(__name__)


def map_error(exc: Exception, default: Code, volume: bool = False) -> Code:
    if isinstance(exc, xerrors.ContainerExisted):
        return Code.CONTAINER_ALREADY_EXIST
    if isinstance(exc, xerrors.VolumeExisted):
        return Code.VOLUME_EXISTED
    if isinstance(exc, xerrors.GpuNotEnough):
        return Code.CONTAINER_GPU_NOT_ENOUGH
    if isinstance(exc, xerrors.CpuNotEnough):
        return Code.CONTAINER_CPU_NOT_ENOUGH
    if isinstance(exc, xerrors.PortNotEnough):
        return Code.CONTAINER_PORT_NOT_ENOUGH
    if isinstance(exc, xerrors.NoPatchRequired):
        return Code.VOLUME_SIZE_NO_NEED_PATCH if volume else Code.CONTAINER_NO_NEED_PATCH
    if isinstance(exc, xerrors.NoRollbackRequired):
        return Code.CONTAINER_NO_NEED_ROLLBACK
    if isinstance(exc, xerrors.VolumeSizeUsedGreaterThanReduced):
        return Code.VOLUME_SIZE_USED_GT_REDUCE
    if isinstance(exc, UnsupportedSizeUnit):
        return Code.VOLUME_SIZE_UNIT if volume else Code.CONTAINER_MEMORY_UNIT
    return default


def log_error(op: str, exc: Exception) -> None:
    log.error("%s failed: %s", op, exc, exc_info=not isinstance(exc, xerrors.GdaError))
