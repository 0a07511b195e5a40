"""dockerd-compatible Engine API server over a real unix socket.

The reference's entire L1/L2 is a live Docker engine
(/root/reference/internal/docker/client.go:8-14); this image has no dockerd
binary and no network to fetch one, so this module serves the Engine API
subset the daemon's DockerRuntime speaks (the ~15 endpoints the reference
uses through the moby client) over a REAL unix socket, backed by the
ProcRuntime: containers are real supervised processes with real rootfs
directories, `GraphDriver.Data.UpperDir` points at the real writable layer
(so rolling-replace migration does real IO through the engine API), exec
runs real processes and returns the real multiplexed stream, commit
snapshots the rootfs, volumes are real directories.

Used by tests/test_live_integration.py (out-of-process, over the socket —
upgrading round 1's in-process fake-transport coverage) and as a dev-mode
engine: ``python -m gpu_docker_api_amd.testing.dockerd_sim --socket /tmp/d.sock``.
"""
from __future__ import annotations

import argparse
import asyncio
import struct
from typing import Any, Dict

from aiohttp import web

from ..models.etcd import ContainerSpec
from ..runtime.proc import ProcRuntime
from ..xerrors import ContainerExisted, ContainerNotExist, VolumeExisted


class DockerdSim:
    def __init__(self, data_dir: str) -> None:
        # resolve GPU UUIDs -> indices for ROCR_VISIBLE_DEVICES like the
        # daemon's proc runtime does; absent GPUs (CPU test boxes) resolve
        # to None and the raw uuid passes through harmlessly
        resolver = None
        try:
            import os as _os

            if not _os.path.exists("/dev/kfd"):
                raise RuntimeError("no GPU on this host")
            from ..parallel.inventory import AmdSmiInventory

            inv = AmdSmiInventory()
            gpus = {g.uuid: g for g in inv.enumerate()}
            if gpus:
                resolver = lambda u: gpus.get(u)  # noqa: E731
        except Exception:
            resolver = None
        self.rt = ProcRuntime(
            base_dir=data_dir, use_cgroups=False, gpu_resolver=resolver or (lambda _u: None)
        )
        # raw HostConfig per container, echoed back by inspect so clients
        # can verify device entries materialize
        self.host_configs: Dict[str, Dict[str, Any]] = {}
        self.execs: Dict[str, Dict[str, Any]] = {}
        self._exec_seq = 0

    # ------------------------------------------------------------- helpers
    @staticmethod
    def _err(status: int, message: str) -> web.Response:
        return web.json_response({"message": message}, status=status)

    def _spec_from_body(self, name: str, body: Dict[str, Any]) -> ContainerSpec:
        spec = ContainerSpec()
        spec.container_name = name
        spec.config = {k: v for k, v in body.items() if k not in ("HostConfig", "NetworkingConfig")}
        spec.host_config = dict(body.get("HostConfig") or {})
        return spec

    def _inspect_json(self, name: str, st) -> Dict[str, Any]:
        hc = dict(self.host_configs.get(name) or {})
        return {
            "Id": st.id,
            "Name": f"/{st.name}",
            "Config": {"Image": st.image, "Env": list(st.env)},
            "HostConfig": {
                **hc,
                "CpusetCpus": st.cpuset_cpus,
                "Memory": st.memory,
                "PortBindings": st.port_bindings,
                "Binds": st.binds,
            },
            "State": {
                "Running": st.running,
                "Paused": st.paused,
                "Status": st.status,
                "Pid": st.pid,
            },
            # the real writable layer: migration through the engine API does
            # real directory IO, as overlay2's UpperDir would
            "GraphDriver": {"Name": "procfs-sim", "Data": {"UpperDir": st.upper_dir}},
        }

    # ------------------------------------------------------------ handlers
    async def create(self, request: web.Request) -> web.Response:
        name = request.query.get("name", "")
        if not name:
            return self._err(400, "container name required")
        body = await request.json()
        spec = self._spec_from_body(name, body)
        try:
            cid = await self.rt.create(spec)
        except ContainerExisted:
            return self._err(409, f"Conflict: {name} already in use")
        except ValueError as exc:  # unsafe name (path traversal guard)
            return self._err(400, str(exc))
        self.host_configs[name] = dict(body.get("HostConfig") or {})
        return web.json_response({"Id": cid, "Warnings": []}, status=201)

    async def lifecycle(self, request: web.Request) -> web.Response:
        name = request.match_info["name"]
        verb = request.match_info["verb"]
        t = int(request.query.get("t", "10"))
        try:
            if verb == "start":
                st = await self.rt.inspect(name)
                if st is not None and st.running:
                    return web.Response(status=304)
                await self.rt.start(name)
            elif verb == "stop":
                st = await self.rt.inspect(name)
                if st is not None and not st.running:
                    return web.Response(status=304)
                await self.rt.stop(name, timeout=t)
            elif verb == "restart":
                await self.rt.restart(name, timeout=t)
            elif verb == "pause":
                await self.rt.pause(name)
            elif verb == "unpause":
                await self.rt.unpause(name)
            else:
                return self._err(404, f"unknown verb {verb}")
        except ContainerNotExist:
            return self._err(404, f"No such container: {name}")
        return web.Response(status=204)

    async def remove(self, request: web.Request) -> web.Response:
        name = request.match_info["name"]
        force = request.query.get("force", "false") == "true"
        try:
            await self.rt.remove(name, force=force)
        except ContainerNotExist:
            return self._err(404, f"No such container: {name}")
        except RuntimeError as exc:
            return self._err(409, str(exc))
        self.host_configs.pop(name, None)
        return web.Response(status=204)

    async def inspect(self, request: web.Request) -> web.Response:
        name = request.match_info["name"]
        st = await self.rt.inspect(name)
        if st is None:
            return self._err(404, f"No such container: {name}")
        return web.json_response(self._inspect_json(name, st))

    async def list_(self, request: web.Request) -> web.Response:
        all_ = request.query.get("all", "false") == "true"
        items = []
        for st in await self.rt.list(all=all_):
            items.append(
                {
                    "Id": st.id,
                    "Names": [f"/{st.name}"],
                    "Image": st.image,
                    "State": st.status if st.status != "exited" else "exited",
                    "Status": st.status,
                }
            )
        return web.json_response(items)

    async def exec_create(self, request: web.Request) -> web.Response:
        name = request.match_info["name"]
        if await self.rt.inspect(name) is None:
            return self._err(404, f"No such container: {name}")
        body = await request.json()
        self._exec_seq += 1
        exec_id = f"exec{self._exec_seq:08d}"
        self.execs[exec_id] = {
            "container": name,
            "cmd": body.get("Cmd") or [],
            "workdir": body.get("WorkingDir", ""),
            "exit_code": None,
        }
        return web.json_response({"Id": exec_id}, status=201)

    async def exec_start(self, request: web.Request) -> web.Response:
        exec_id = request.match_info["id"]
        e = self.execs.get(exec_id)
        if e is None:
            return self._err(404, f"No such exec: {exec_id}")
        try:
            out, rc = await self.rt.execute_rc(e["container"], e["cmd"], e["workdir"])
        except (ContainerNotExist, RuntimeError) as exc:
            return self._err(409, str(exc))
        e["exit_code"] = rc
        payload = out.encode()
        frame = bytes([1, 0, 0, 0]) + struct.pack(">I", len(payload)) + payload
        return web.Response(body=frame, content_type="application/vnd.docker.raw-stream")

    async def exec_json(self, request: web.Request) -> web.Response:
        e = self.execs.get(request.match_info["id"])
        if e is None:
            return self._err(404, "no such exec")
        return web.json_response({"ExitCode": e["exit_code"] or 0, "Running": False})

    async def container_logs(self, request: web.Request) -> web.Response:
        name = request.match_info["name"]
        tail = int(request.query.get("tail", "200"))
        try:
            out = await self.rt.logs(name, tail=tail)
        except ContainerNotExist:
            return self._err(404, f"No such container: {name}")
        payload = out.encode()
        frame = bytes([1, 0, 0, 0]) + struct.pack(">I", len(payload)) + payload
        return web.Response(body=frame, content_type="application/vnd.docker.raw-stream")

    async def container_stats(self, request: web.Request) -> web.Response:
        name = request.match_info["name"]
        try:
            st = await self.rt.stats(name)
        except ContainerNotExist:
            return self._err(404, f"No such container: {name}")
        return web.json_response(
            {
                "cpu_stats": {"cpu_usage": {"total_usage": int(st["cpuSeconds"] * 1e9)}},
                "memory_stats": {"usage": st["memoryBytes"]},
                "pids_stats": {"current": st["pids"]},
            }
        )

    async def commit(self, request: web.Request) -> web.Response:
        name = request.query.get("container", "")
        repo = request.query.get("repo", "")
        tag = request.query.get("tag", "")
        try:
            ref = await self.rt.commit(name, repo, tag)
        except ContainerNotExist:
            return self._err(404, f"No such container: {name}")
        return web.json_response({"Id": f"sha256:{abs(hash(ref)) :x}"}, status=201)

    async def volume_create(self, request: web.Request) -> web.Response:
        body = await request.json()
        name = body.get("Name", "")
        try:
            vs = await self.rt.volume_create(name, body.get("DriverOpts") or None)
        except VolumeExisted:
            return self._err(409, f"volume {name} exists")
        except ValueError as exc:
            return self._err(400, str(exc))
        return web.json_response(
            {"Name": vs.name, "Mountpoint": vs.mountpoint, "Driver": "local",
             "Options": vs.options}, status=201)

    async def volume_inspect(self, request: web.Request) -> web.Response:
        vs = await self.rt.volume_inspect(request.match_info["name"])
        if vs is None:
            return self._err(404, "no such volume")
        return web.json_response(
            {"Name": vs.name, "Mountpoint": vs.mountpoint, "Driver": "local",
             "Options": vs.options})

    async def volume_remove(self, request: web.Request) -> web.Response:
        name = request.match_info["name"]
        if await self.rt.volume_inspect(name) is None:
            return self._err(404, "no such volume")
        await self.rt.volume_remove(name)
        return web.Response(status=204)

    async def ping(self, request: web.Request) -> web.Response:
        return web.Response(text="OK")

    def app(self) -> web.Application:
        app = web.Application()
        r = app.router
        # {v} swallows the API-version prefix (v1.41 etc.)
        r.add_post("/{v}/containers/create", self.create)
        r.add_get("/{v}/containers/json", self.list_)
        r.add_get("/{v}/containers/{name}/json", self.inspect)
        r.add_get("/{v}/containers/{name}/logs", self.container_logs)
        r.add_get("/{v}/containers/{name}/stats", self.container_stats)
        r.add_post("/{v}/containers/{name}/exec", self.exec_create)
        r.add_post("/{v}/containers/{name}/{verb}", self.lifecycle)
        r.add_delete("/{v}/containers/{name}", self.remove)
        r.add_post("/{v}/exec/{id}/start", self.exec_start)
        r.add_get("/{v}/exec/{id}/json", self.exec_json)
        r.add_post("/{v}/commit", self.commit)
        r.add_post("/{v}/volumes/create", self.volume_create)
        r.add_get("/{v}/volumes/{name}", self.volume_inspect)
        r.add_delete("/{v}/volumes/{name}", self.volume_remove)
        r.add_get("/{v}/_ping", self.ping)
        r.add_get("/_ping", self.ping)
        return app

    async def close(self) -> None:
        await self.rt.close()


async def serve(socket_path: str, data_dir: str) -> None:
    sim = DockerdSim(data_dir)
    runner = web.AppRunner(sim.app())
    await runner.setup()
    site = web.UnixSite(runner, socket_path)
    await site.start()
    print(f"dockerd-sim listening on {socket_path}", flush=True)
    try:
        while True:
            await asyncio.sleep(3600)
    finally:
        await runner.cleanup()
        await sim.close()


def main() -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--socket", required=True, help="unix socket path to serve")
    p.add_argument("--data", required=True, help="runtime data directory")
    args = p.parse_args()
    asyncio.run(serve(args.socket, args.data))


if __name__ == "__main__":
    main()
