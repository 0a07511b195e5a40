"""Container runtime driver interface.

The reference can only drive dockerd (moby client singleton,
/root/reference/internal/docker/client.go). Here runtimes are pluggable:

* ``docker`` — Docker Engine API over the unix socket, with ROCm device
  injection (/dev/kfd + /dev/dri/renderD*) replacing the reference's
  ``DeviceRequests{Driver:"cdi"}`` + ``Runtime:"nvidia"``
  (services/replicaset_nomock.go:128-140);
* ``proc`` — a native lightweight runtime supervising real host processes
  (rootfs dirs, cgroup-v2 limits when permitted, GPU isolation via
  ``ROCR_VISIBLE_DEVICES``) — what dockerd-less GPU nodes and bench.py use;
* ``mock`` — in-process fake (the reference's ``-tags mock`` flavor as a
  config choice, replicaset_mock.go:44-60).

Drivers consume the stored :class:`ContainerSpec` directly (its
config/hostConfig are Docker-Engine-shaped), so specs round-trip through the
state store unchanged regardless of driver.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from ..models.etcd import ContainerSpec
from ..parallel.inventory import GpuInfo

GpuResolver = Callable[[str], Optional[GpuInfo]]


@dataclass
class ContainerState:
    id: str
    name: str
    image: str = ""
    running: bool = False
    paused: bool = False
    status: str = "created"
    pid: int = 0
    env: List[str] = field(default_factory=list)
    gpu_uuids: List[str] = field(default_factory=list)
    cpuset_cpus: str = ""
    memory: int = 0
    port_bindings: Dict[str, List[Dict[str, str]]] = field(default_factory=dict)
    upper_dir: str = ""  # writable rootfs layer (migration source/target)
    binds: List[str] = field(default_factory=list)
    extra: Dict[str, Any] = field(default_factory=dict)


@dataclass
class VolumeState:
    name: str
    mountpoint: str
    driver: str = "local"
    options: Dict[str, str] = field(default_factory=dict)


class RuntimeDriver:
    """All methods raise on hard failure; inspect returns None when absent."""

    # True when the driver owns the container rootfs directory outright
    # (proc/mock): the control plane may rename it away during replacement.
    # False for docker (UpperDir belongs to overlayfs; must be copied).
    owns_rootfs = False

    async def create(self, spec: ContainerSpec) -> str:
        """Create (not start) a container named spec.container_name; returns id."""
        raise NotImplementedError

    async def start(self, name: str) -> None:
        raise NotImplementedError

    async def stop(self, name: str, timeout: int = 10) -> None:
        raise NotImplementedError

    async def pause(self, name: str) -> None:
        raise NotImplementedError

    async def unpause(self, name: str) -> None:
        raise NotImplementedError

    async def restart(self, name: str, timeout: int = 10) -> None:
        raise NotImplementedError

    async def remove(self, name: str, force: bool = True) -> None:
        raise NotImplementedError

    async def inspect(self, name: str) -> Optional[ContainerState]:
        raise NotImplementedError

    async def list(self, all: bool = True) -> List[ContainerState]:
        raise NotImplementedError

    async def execute(self, name: str, cmd: List[str], workdir: str = "") -> str:
        """Run a command inside the container, return combined output."""
        out, _rc = await self.execute_rc(name, cmd, workdir)
        return out

    async def execute_rc(self, name: str, cmd: List[str], workdir: str = ""):
        """Run a command; returns (combined output, exit code)."""
        raise NotImplementedError

    async def commit(self, name: str, image: str, tag: str = "") -> str:
        raise NotImplementedError

    async def logs(self, name: str, tail: int = 200) -> str:
        """Captured container output (extension: the reference exposes no
        logs route; its users must go to dockerd directly)."""
        raise NotImplementedError

    async def stats(self, name: str) -> Dict:
        """Live resource usage {cpuSeconds, memoryBytes, pids} (extension:
        the reference has no stats route)."""
        raise NotImplementedError

    async def image_import(self, ref: str, src_path: str) -> str:
        """Register a local directory as image ``ref`` (proc/mock: the
        'pull it locally first' step the reference delegates to dockerd).
        Drivers whose engine owns images raise RuntimeError."""
        raise NotImplementedError

    async def image_list(self) -> List[Dict]:
        raise NotImplementedError

    # ---- volumes ----
    async def volume_create(
        self, name: str, driver_opts: Optional[Dict[str, str]] = None
    ) -> VolumeState:
        raise NotImplementedError

    async def volume_remove(self, name: str, force: bool = True) -> None:
        raise NotImplementedError

    async def volume_inspect(self, name: str) -> Optional[VolumeState]:
        raise NotImplementedError

    async def close(self) -> None:
        pass
