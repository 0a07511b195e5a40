"""State-store key scheme.

Identical to the reference so an etcd populated by either implementation is
readable by the other (/root/reference/internal/etcd/common.go:15-30):

    /gpu-docker-api/apis/v1/<resource>/<key>
"""
from __future__ import annotations

import enum

PREFIX = "/gpu-docker-api"
API_VERSION = "apis/v1"


class Resource(str, enum.Enum):
    CONTAINERS = "containers"
    VOLUMES = "volumes"
    VERSIONS = "versions"
    MERGES = "merges"
    GPUS = "gpus"
    CPUS = "cpus"
    PORTS = "ports"


def resource_prefix(resource: Resource | str) -> str:
    r = resource.value if isinstance(resource, Resource) else resource
    return f"{PREFIX}/{API_VERSION}/{r}"


def resource_key(resource: Resource | str, key: str) -> str:
    return f"{resource_prefix(resource)}/{key}"


CONTAINER_PREFIX = resource_prefix(Resource.CONTAINERS)
VOLUME_PREFIX = resource_prefix(Resource.VOLUMES)
VERSIONS = resource_prefix(Resource.VERSIONS)
MERGES = resource_prefix(Resource.MERGES)
GPUS = resource_prefix(Resource.GPUS)
CPUS = resource_prefix(Resource.CPUS)
PORTS = resource_prefix(Resource.PORTS)

# Singleton map keys (reference: gpuscheduler.go:21, cpuscheduler.go:19,
# portscheduler.go:20, version/version.go:21-22, version/merge.go:13).
GPU_STATUS_MAP_KEY = "gpuStatusMapKey"
CPU_STATUS_MAP_KEY = "cpuStatusMapKey"
USED_PORT_SET_KEY = "usedPortSetKey"
CONTAINER_VERSION_MAP_KEY = "containerVersionMapKey"
VOLUME_VERSION_MAP_KEY = "volumeVersionMapKey"
CONTAINER_MERGE_MAP_KEY = "containerMergeMapKey"
RELEASED_SET_KEY = "releasedSetKey"
