from .container import (
    Bind,
    ContainerCommit,
    ContainerExecute,
    ContainerRun,
    CpuPatch,
    GpuPatch,
    MemoryPatch,
    PatchRequest,
    RollbackRequest,
    VolumePatch,
)
from .volume import VolumeCreate, VolumeSize
from .etcd import ContainerSpec, VolumeSpec, HistoryItem
from .memory import MEMORY_UNITS, to_bytes, parse_size

__all__ = [
    "Bind",
    "ContainerCommit",
    "ContainerExecute",
    "ContainerRun",
    "CpuPatch",
    "GpuPatch",
    "MemoryPatch",
    "PatchRequest",
    "RollbackRequest",
    "VolumePatch",
    "VolumeCreate",
    "VolumeSize",
    "ContainerSpec",
    "VolumeSpec",
    "HistoryItem",
    "MEMORY_UNITS",
    "to_bytes",
    "parse_size",
]
