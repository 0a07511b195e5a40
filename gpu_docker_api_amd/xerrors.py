"""Sentinel error types.

The reference uses string-sentinel constructor/predicate pairs
(/root/reference/internal/xerrors/*.go). Python exceptions subsume both
roles: each condition is a distinct exception class, and ``isinstance``
is the predicate. Routers map these to the numeric business codes in
``routers/codes.py``.
"""
from __future__ import annotations


class GdaError(Exception):
    """Base class for all control-plane errors."""


# -- container (reference: internal/xerrors/container.go) --------------------
class ContainerExisted(GdaError):
    pass


class ContainerNotExist(GdaError):
    pass


# -- volume (reference: internal/xerrors/volume.go) --------------------------
class VolumeExisted(GdaError):
    pass


class VolumeSizeUsedGreaterThanReduced(GdaError):
    """Shrinking a volume below its currently used bytes."""


# -- schedulers (reference: internal/xerrors/scheduler.go) -------------------
class GpuNotEnough(GdaError):
    pass


class CpuNotEnough(GdaError):
    pass


class PortNotEnough(GdaError):
    pass


# -- state store (reference: internal/xerrors/etcd.go) -----------------------
class NotExistInStore(GdaError):
    """Key absent from the state store (reference: xerrors.NotExistInEtcdError)."""


class RevisionCompacted(GdaError):
    """Requested revision is older than the store's compaction point."""


# -- services (reference: internal/xerrors/common.go) ------------------------
class NoPatchRequired(GdaError):
    pass


class NoRollbackRequired(GdaError):
    pass


class RuntimeUnavailable(GdaError):
    """The container runtime driver cannot be reached."""


class NativeOpUnavailable(GdaError):
    """A native extension (.so) is required but not built/loadable."""
