from .keys import (
    CONTAINER_PREFIX,
    CPUS,
    GPUS,
    MERGES,
    PORTS,
    VERSIONS,
    VOLUME_PREFIX,
    Resource,
    resource_prefix,
    resource_key,
)
from .mvcc import KeyValue, MemoryMVCC
from .store import StateStore, MemoryStore
from .workqueue import WorkQueue, PutKeyValue, DelKey

__all__ = [
    "CONTAINER_PREFIX",
    "CPUS",
    "GPUS",
    "MERGES",
    "PORTS",
    "VERSIONS",
    "VOLUME_PREFIX",
    "Resource",
    "resource_prefix",
    "resource_key",
    "KeyValue",
    "MemoryMVCC",
    "StateStore",
    "MemoryStore",
    "WorkQueue",
    "PutKeyValue",
    "DelKey",
]
