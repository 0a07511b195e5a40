"""Docker Engine API driver.

Speaks the Engine HTTP API directly over the unix socket (aiohttp) — the ~15
endpoints the reference uses through the moby client
(/root/reference/internal/docker/client.go, services/*.go). GPU attachment is
ROCm-native: ``/dev/kfd`` + per-GPU ``/dev/dri/renderD*`` device entries and
video/render group-add on the stock runc runtime (see devices.py), replacing
the reference's nvidia runtime + CDI DeviceRequests.
"""
from __future__ import annotations

import json
import struct
from typing import Any, Dict, List, Optional

import aiohttp

from ..models.etcd import ContainerSpec
from ..xerrors import ContainerExisted, ContainerNotExist, RuntimeUnavailable, VolumeExisted
from .base import ContainerState, GpuResolver, RuntimeDriver, VolumeState
from .devices import docker_devices_for, docker_group_add


class DockerRuntime(RuntimeDriver):
    def __init__(
        self,
        socket_path: str = "/var/run/docker.sock",
        gpu_resolver: Optional[GpuResolver] = None,
        api_version: str = "v1.41",
        use_cdi: bool = False,
    ) -> None:
        self.socket_path = socket_path
        self.api = api_version
        self.gpu_resolver: GpuResolver = gpu_resolver or (lambda _u: None)
        # CDI mode: emit amd.com/gpu=<index> DeviceRequests for hosts with
        # CDI specs configured (docker >= 25 / podman); default is direct
        # /dev/kfd + /dev/dri injection which needs no host-side spec.
        self.use_cdi = use_cdi
        self._session: Optional[aiohttp.ClientSession] = None

    def _sess(self) -> aiohttp.ClientSession:
        if self._session is None or self._session.closed:
            conn = aiohttp.UnixConnector(path=self.socket_path)
            self._session = aiohttp.ClientSession(
                connector=conn, timeout=aiohttp.ClientTimeout(total=120)
            )
        return self._session

    async def _call(
        self,
        method: str,
        path: str,
        body: Optional[Dict[str, Any]] = None,
        params: Optional[Dict[str, str]] = None,
        ok: tuple = (200, 201, 204),
        raw: bool = False,
    ):
        url = f"http://localhost/{self.api}{path}"
        try:
            async with self._sess().request(method, url, json=body, params=params) as resp:
                data = await resp.read()
                if resp.status == 404:
                    raise ContainerNotExist(path)
                if resp.status == 409 and path.endswith("/create"):
                    # name conflict on create; elsewhere 409 means a state
                    # conflict (e.g. exec on a stopped container) and falls
                    # through to the generic error below
                    raise ContainerExisted(path)
                if resp.status not in ok:
                    msg = data.decode(errors="replace")
                    raise RuntimeError(f"docker {method} {path}: {resp.status} {msg}")
                if raw:
                    return data
                return json.loads(data) if data else None
        except aiohttp.ClientConnectionError as exc:
            raise RuntimeUnavailable(f"dockerd unreachable at {self.socket_path}: {exc}") from exc

    # ------------------------------------------------------------ containers
    def _materialize_host_config(self, spec: ContainerSpec) -> Dict[str, Any]:
        hc = dict(spec.host_config)
        uuids = spec.gpu_uuids
        if self.use_cdi and uuids:
            ids = []
            for u in uuids:
                info = self.gpu_resolver(u)
                ids.append(f"amd.com/gpu={info.index if info else u}")
            hc["DeviceRequests"] = [{"Driver": "cdi", "DeviceIDs": ids}]
            hc.pop("Runtime", None)
            return hc
        devices = docker_devices_for(uuids, self.gpu_resolver)
        if devices:
            existing = {d.get("PathOnHost") for d in (hc.get("Devices") or [])}
            hc["Devices"] = (hc.get("Devices") or []) + [
                d for d in devices if d["PathOnHost"] not in existing
            ]
            groups = set(hc.get("GroupAdd") or [])
            hc["GroupAdd"] = sorted(groups | set(docker_group_add()))
        # no NVIDIA runtime, no CDI DeviceRequests: plain runc
        hc.pop("Runtime", None)
        hc.pop("DeviceRequests", None)
        return hc

    async def create(self, spec: ContainerSpec) -> str:
        body = {
            **spec.config,
            "HostConfig": self._materialize_host_config(spec),
        }
        if spec.networking_config:
            body["NetworkingConfig"] = spec.networking_config
        try:
            data = await self._call(
                "POST", "/containers/create", body=body, params={"name": spec.container_name}
            )
        except ContainerExisted:
            raise ContainerExisted(spec.container_name) from None
        return data["Id"]

    async def start(self, name: str) -> None:
        await self._call("POST", f"/containers/{name}/start", ok=(204, 304))

    async def stop(self, name: str, timeout: int = 10) -> None:
        await self._call(
            "POST", f"/containers/{name}/stop", params={"t": str(timeout)}, ok=(204, 304)
        )

    async def pause(self, name: str) -> None:
        await self._call("POST", f"/containers/{name}/pause", ok=(204,))

    async def unpause(self, name: str) -> None:
        await self._call("POST", f"/containers/{name}/unpause", ok=(204,))

    async def restart(self, name: str, timeout: int = 10) -> None:
        await self._call(
            "POST", f"/containers/{name}/restart", params={"t": str(timeout)}, ok=(204,)
        )

    async def remove(self, name: str, force: bool = True) -> None:
        await self._call(
            "DELETE",
            f"/containers/{name}",
            params={"force": "true" if force else "false", "v": "false"},
            ok=(204,),
        )

    @staticmethod
    def _state_from_inspect(d: Dict[str, Any]) -> ContainerState:
        cfg = d.get("Config") or {}
        hc = d.get("HostConfig") or {}
        state = d.get("State") or {}
        env = cfg.get("Env") or []
        gpu_uuids: List[str] = []
        for e in env:
            if e.startswith("GDA_GPU_UUIDS="):
                gpu_uuids = [u for u in e.split("=", 1)[1].split(",") if u]
        return ContainerState(
            id=d.get("Id", ""),
            name=(d.get("Name") or "").lstrip("/"),
            image=cfg.get("Image", ""),
            running=bool(state.get("Running")),
            paused=bool(state.get("Paused")),
            status=state.get("Status", ""),
            pid=int(state.get("Pid") or 0),
            env=env,
            gpu_uuids=gpu_uuids,
            cpuset_cpus=hc.get("CpusetCpus", "") or "",
            memory=int(hc.get("Memory") or 0),
            port_bindings=hc.get("PortBindings") or {},
            upper_dir=((d.get("GraphDriver") or {}).get("Data") or {}).get("UpperDir", ""),
            binds=hc.get("Binds") or [],
            extra={"GraphDriver": d.get("GraphDriver")},
        )

    async def inspect(self, name: str) -> Optional[ContainerState]:
        try:
            d = await self._call("GET", f"/containers/{name}/json")
        except ContainerNotExist:
            return None
        return self._state_from_inspect(d)

    async def list(self, all: bool = True) -> List[ContainerState]:
        items = await self._call(
            "GET", "/containers/json", params={"all": "true" if all else "false"}
        )
        out = []
        for it in items or []:
            out.append(
                ContainerState(
                    id=it.get("Id", ""),
                    name=(it.get("Names") or ["/"])[0].lstrip("/"),
                    image=it.get("Image", ""),
                    running=it.get("State") == "running",
                    paused=it.get("State") == "paused",
                    status=it.get("Status", ""),
                )
            )
        return out

    @staticmethod
    def _demux_stream(data: bytes) -> str:
        """Decode docker's multiplexed attach stream (8-byte frame headers)."""
        out = []
        i = 0
        while i + 8 <= len(data):
            _stream, size = data[i], struct.unpack(">I", data[i + 4 : i + 8])[0]
            out.append(data[i + 8 : i + 8 + size])
            i += 8 + size
        if not out and data:
            return data.decode(errors="replace")  # TTY mode: raw
        return b"".join(out).decode(errors="replace")

    async def execute_rc(self, name: str, cmd: List[str], workdir: str = ""):
        body: Dict[str, Any] = {
            "AttachStdout": True,
            "AttachStderr": True,
            "Cmd": cmd,
        }
        if workdir:
            body["WorkingDir"] = workdir
        data = await self._call("POST", f"/containers/{name}/exec", body=body)
        exec_id = data["Id"]
        raw = await self._call(
            "POST", f"/exec/{exec_id}/start", body={"Detach": False, "Tty": False}, raw=True
        )
        rc = 0
        try:
            info = await self._call("GET", f"/exec/{exec_id}/json")
            rc = int(info.get("ExitCode") or 0)
        except Exception:  # older daemons: treat as success
            rc = 0
        return self._demux_stream(raw or b""), rc

    async def logs(self, name: str, tail: int = 200) -> str:
        raw = await self._call(
            "GET",
            f"/containers/{name}/logs",
            params={"stdout": "true", "stderr": "true", "tail": str(tail)},
            raw=True,
        )
        return self._demux_stream(raw or b"")

    async def image_import(self, ref: str, src_path: str) -> str:
        raise RuntimeError(
            "the docker engine owns its image store: docker pull/load there"
        )

    async def image_list(self) -> List[Dict[str, Any]]:
        items = await self._call("GET", "/images/json") or []
        return [
            {"ref": (it.get("RepoTags") or [it.get("Id", "")])[0], "id": it.get("Id", "")}
            for it in items
        ]

    async def stats(self, name: str) -> Dict[str, Any]:
        d = await self._call(
            "GET", f"/containers/{name}/stats",
            params={"stream": "false", "one-shot": "true"},
        )
        cpu_ns = ((d.get("cpu_stats") or {}).get("cpu_usage") or {}).get("total_usage", 0)
        mem = (d.get("memory_stats") or {}).get("usage", 0)
        pids = ((d.get("pids_stats") or {}).get("current")) or 0
        return {
            "running": True,
            "cpuSeconds": round(int(cpu_ns) / 1e9, 3),
            "memoryBytes": int(mem),
            "pids": int(pids),
        }

    async def commit(self, name: str, image: str, tag: str = "") -> str:
        repo, _, t = image.partition(":")
        if not t:
            t = tag
        await self._call(
            "POST",
            "/commit",
            body={},
            params={"container": name, "repo": repo, **({"tag": t} if t else {})},
        )
        return f"{repo}:{t}" if t else repo

    # --------------------------------------------------------------- volumes
    async def volume_create(
        self, name: str, driver_opts: Optional[Dict[str, str]] = None
    ) -> VolumeState:
        body = {"Name": name, "Driver": "local"}
        if driver_opts:
            body["DriverOpts"] = driver_opts
        try:
            d = await self._call("POST", "/volumes/create", body=body)
        except ContainerExisted:
            raise VolumeExisted(name) from None
        return VolumeState(
            name=d.get("Name", name),
            mountpoint=d.get("Mountpoint", ""),
            driver=d.get("Driver", "local"),
            options=d.get("Options") or {},
        )

    async def volume_remove(self, name: str, force: bool = True) -> None:
        await self._call(
            "DELETE",
            f"/volumes/{name}",
            params={"force": "true" if force else "false"},
            ok=(204,),
        )

    async def volume_inspect(self, name: str) -> Optional[VolumeState]:
        try:
            d = await self._call("GET", f"/volumes/{name}")
        except ContainerNotExist:
            return None
        return VolumeState(
            name=d.get("Name", name),
            mountpoint=d.get("Mountpoint", ""),
            driver=d.get("Driver", "local"),
            options=d.get("Options") or {},
        )

    async def close(self) -> None:
        if self._session and not self._session.closed:
            await self._session.close()
