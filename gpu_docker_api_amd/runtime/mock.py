"""In-process fake runtime.

The config-selected equivalent of the reference's compile-time mock flavor
(/root/reference/internal/services/replicaset_mock.go,
gpuscheduler_mock.go): full control-plane machinery with no dockerd and no
GPUs. Unlike the reference's mock (which only strips DeviceRequests), this
one backs every container with a real rootfs directory and every volume with
a real mountpoint directory, so the rolling-replace data-migration paths run
for real in unit tests.
"""
from __future__ import annotations

import asyncio
import os
import shutil
import subprocess
import uuid as uuidlib
from typing import Dict, List, Optional

from ..models.etcd import ContainerSpec
from ..xerrors import ContainerExisted, ContainerNotExist, VolumeExisted
from .base import ContainerState, RuntimeDriver, VolumeState


class MockRuntime(RuntimeDriver):
    owns_rootfs = True

    def __init__(self, base_dir: str = "", fail_on: Optional[set] = None) -> None:
        self.base = base_dir or os.path.join(os.getcwd(), ".state", "mockrt")
        os.makedirs(os.path.join(self.base, "containers"), exist_ok=True)
        os.makedirs(os.path.join(self.base, "volumes"), exist_ok=True)
        self.containers: Dict[str, ContainerState] = {}
        self.volumes: Dict[str, VolumeState] = {}
        self.images: Dict[str, str] = {}  # image name -> seed dir (optional)
        self._logs: Dict[str, str] = {}   # synthetic console output
        # test hook: operation names that should raise (failure-injection)
        self.fail_on = fail_on or set()

    def _maybe_fail(self, op: str) -> None:
        if op in self.fail_on:
            raise RuntimeError(f"injected failure: {op}")

    # ------------------------------------------------------------ containers
    async def create(self, spec: ContainerSpec) -> str:
        self._maybe_fail("create")
        name = spec.container_name
        if name in self.containers:
            raise ContainerExisted(name)
        cid = uuidlib.uuid4().hex[:12]
        rootfs = os.path.join(self.base, "containers", name, "rootfs")
        os.makedirs(rootfs, exist_ok=True)
        seed = self.images.get(spec.image)
        if seed and os.path.isdir(seed):
            shutil.copytree(seed, rootfs, dirs_exist_ok=True)
        st = ContainerState(
            id=cid,
            name=name,
            image=spec.image,
            running=False,
            status="created",
            env=list(spec.env),
            gpu_uuids=list(spec.gpu_uuids),
            cpuset_cpus=spec.cpuset_cpus,
            memory=spec.memory_bytes,
            port_bindings=dict(spec.host_config.get("PortBindings") or {}),
            upper_dir=rootfs,
            binds=list(spec.host_config.get("Binds") or []),
        )
        self.containers[name] = st
        return cid

    def _get(self, name: str) -> ContainerState:
        if name not in self.containers:
            raise ContainerNotExist(name)
        return self.containers[name]

    async def start(self, name: str) -> None:
        self._maybe_fail("start")
        st = self._get(name)
        st.running, st.paused, st.status = True, False, "running"

    async def stop(self, name: str, timeout: int = 10) -> None:
        self._maybe_fail("stop")
        st = self._get(name)
        st.running, st.paused, st.status = False, False, "exited"

    async def pause(self, name: str) -> None:
        st = self._get(name)
        st.paused, st.status = True, "paused"

    async def unpause(self, name: str) -> None:
        st = self._get(name)
        st.paused, st.status = False, "running"
        st.running = True

    async def restart(self, name: str, timeout: int = 10) -> None:
        await self.stop(name, timeout)
        await self.start(name)

    async def remove(self, name: str, force: bool = True) -> None:
        self._maybe_fail("remove")
        st = self._get(name)
        if st.running and not force:
            raise RuntimeError(f"{name} is running")
        self.containers.pop(name, None)
        shutil.rmtree(os.path.join(self.base, "containers", name), ignore_errors=True)

    async def inspect(self, name: str) -> Optional[ContainerState]:
        return self.containers.get(name)

    async def list(self, all: bool = True) -> List[ContainerState]:
        return [c for c in self.containers.values() if all or c.running]

    async def execute_rc(self, name: str, cmd: List[str], workdir: str = ""):
        """Run the command on the host chrooted-by-cwd into the rootfs dir
        (close enough for tests; proc/docker drivers do it for real)."""
        self._maybe_fail("execute")
        st = self._get(name)
        if not st.running:
            raise RuntimeError(f"{name} is not running")
        cwd = os.path.join(st.upper_dir, workdir.lstrip("/")) if workdir else st.upper_dir
        os.makedirs(cwd, exist_ok=True)
        proc = await asyncio.create_subprocess_exec(
            *cmd,
            cwd=cwd,
            stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT,
        )
        out, _ = await proc.communicate()
        return out.decode(errors="replace"), proc.returncode

    async def logs(self, name: str, tail: int = 200) -> str:
        if name not in self.containers:
            raise ContainerNotExist(name)
        return self._logs.get(name, "")

    async def stats(self, name: str) -> dict:
        if name not in self.containers:
            raise ContainerNotExist(name)
        st = self.containers[name]
        return {"running": st.running, "cpuSeconds": 0.0, "memoryBytes": 0,
                "pids": 0, "restarts": 0}

    async def image_import(self, ref: str, src_path: str) -> str:
        import os as _os

        if not _os.path.isdir(src_path):
            raise FileNotFoundError(src_path)
        self.images[ref] = src_path
        return ref

    async def image_list(self) -> list:
        return [{"ref": r, "path": p} for r, p in sorted(self.images.items())]

    async def commit(self, name: str, image: str, tag: str = "") -> str:
        self._maybe_fail("commit")
        st = self._get(name)
        ref = f"{image}:{tag}" if tag else image
        seed = os.path.join(self.base, "images", ref.replace("/", "_").replace(":", "_"))
        shutil.rmtree(seed, ignore_errors=True)
        shutil.copytree(st.upper_dir, seed)
        self.images[ref] = seed
        if not tag:
            self.images[image] = seed
        return ref

    # --------------------------------------------------------------- volumes
    async def volume_create(
        self, name: str, driver_opts: Optional[Dict[str, str]] = None
    ) -> VolumeState:
        self._maybe_fail("volume_create")
        if name in self.volumes:
            raise VolumeExisted(name)
        mp = os.path.join(self.base, "volumes", name, "_data")
        os.makedirs(mp, exist_ok=True)
        vs = VolumeState(name=name, mountpoint=mp, options=dict(driver_opts or {}))
        self.volumes[name] = vs
        return vs

    async def volume_remove(self, name: str, force: bool = True) -> None:
        self.volumes.pop(name, None)
        shutil.rmtree(os.path.join(self.base, "volumes", name), ignore_errors=True)

    async def volume_inspect(self, name: str) -> Optional[VolumeState]:
        return self.volumes.get(name)
