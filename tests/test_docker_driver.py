"""DockerRuntime against a fake Docker Engine API server on a unix socket:
verifies request construction — ROCm device injection, no NVIDIA runtime,
exec stream demux — without a real dockerd."""
import asyncio
import json
import struct

import pytest
from aiohttp import web

from gpu_docker_api_amd.models.etcd import ContainerSpec
from gpu_docker_api_amd.parallel.inventory import GpuInfo
from gpu_docker_api_amd.runtime.docker import DockerRuntime


class FakeDockerd:
    def __init__(self):
        self.requests = []  # (method, path, body)
        self.containers = {}

    def app(self):
        app = web.Application()
        app.router.add_route("*", "/{tail:.*}", self.handle)
        return app

    async def handle(self, request: web.Request):
        body = None
        if request.can_read_body:
            raw = await request.read()
            if raw:
                try:
                    body = json.loads(raw)
                except json.JSONDecodeError:
                    body = raw
        path = "/" + request.match_info["tail"]
        self.requests.append((request.method, path, body, dict(request.query)))

        if path.endswith("/containers/create"):
            name = request.query.get("name", "noname")
            self.containers[name] = body
            return web.json_response({"Id": "cid123"}, status=201)
        if "/exec/" in path and path.endswith("/start"):
            payload = b"hello-from-exec\n"
            frame = bytes([1, 0, 0, 0]) + struct.pack(">I", len(payload)) + payload
            return web.Response(body=frame, content_type="application/octet-stream")
        if path.endswith("/start") or path.endswith("/stop"):
            return web.Response(status=204)
        if "/containers/" in path and path.endswith("/json"):
            name = path.split("/")[-2]
            if name not in self.containers:
                return web.json_response({"message": "no such container"}, status=404)
            spec = self.containers[name]
            return web.json_response(
                {
                    "Id": "cid123",
                    "Name": f"/{name}",
                    "Config": {"Image": spec.get("Image"), "Env": spec.get("Env", [])},
                    "HostConfig": spec.get("HostConfig", {}),
                    "State": {"Running": True, "Paused": False, "Status": "running", "Pid": 42},
                    "GraphDriver": {"Data": {"UpperDir": "/var/lib/docker/overlay2/x/diff"}},
                }
            )
        if path.endswith("/exec"):
            return web.json_response({"Id": "exec1"}, status=201)
        if "/exec/" in path and path.endswith("/start"):
            payload = b"hello-from-exec\n"
            frame = bytes([1, 0, 0, 0]) + struct.pack(">I", len(payload)) + payload
            return web.Response(body=frame, content_type="application/octet-stream")
        if path.endswith("/volumes/create"):
            return web.json_response(
                {"Name": body["Name"], "Mountpoint": f"/var/lib/docker/volumes/{body['Name']}/_data",
                 "Driver": "local", "Options": body.get("DriverOpts", {})},
                status=201,
            )
        return web.Response(status=204)


def _resolver(uuid):
    idx = int(uuid.split("-")[-1])
    return GpuInfo(
        index=idx,
        uuid=uuid,
        render_node=f"/dev/dri/renderD{128 + idx}",
        card_node=f"/dev/dri/card{idx}",
    )


def test_create_cdi_mode(tmp_path, run):
    async def main():
        fake = FakeDockerd()
        runner = web.AppRunner(fake.app())
        await runner.setup()
        sock = str(tmp_path / "docker.sock")
        await web.UnixSite(runner, sock).start()
        rt = DockerRuntime(socket_path=sock, gpu_resolver=_resolver, use_cdi=True)
        spec = ContainerSpec()
        spec.config = {"Image": "rocm/dev", "Env": []}
        spec.host_config = {}
        spec.container_name = "cdi-1"
        spec.gpu_uuids = ["MockMI355X-2"]
        await rt.create(spec)
        hc = fake.containers["cdi-1"]["HostConfig"]
        assert hc["DeviceRequests"] == [
            {"Driver": "cdi", "DeviceIDs": ["amd.com/gpu=2"]}
        ]
        assert "Devices" not in hc
        await rt.close()
        await runner.cleanup()

    run(main())


def test_create_injects_rocm_devices(tmp_path, run):
    async def main():
        fake = FakeDockerd()
        runner = web.AppRunner(fake.app())
        await runner.setup()
        sock = str(tmp_path / "docker.sock")
        site = web.UnixSite(runner, sock)
        await site.start()

        rt = DockerRuntime(socket_path=sock, gpu_resolver=_resolver)
        spec = ContainerSpec()
        spec.config = {"Image": "rocm/dev", "Env": [], "Cmd": ["sleep", "inf"]}
        spec.host_config = {
            "Binds": ["v:/data"],
            "Runtime": "nvidia",  # must be stripped
            "DeviceRequests": [{"Driver": "cdi"}],  # must be stripped
        }
        spec.container_name = "demo-1"
        spec.gpu_uuids = ["MockMI355X-0", "MockMI355X-3"]
        cid = await rt.create(spec)
        assert cid == "cid123"

        created = fake.containers["demo-1"]
        hc = created["HostConfig"]
        paths = [d["PathOnHost"] for d in hc["Devices"]]
        assert "/dev/kfd" in paths
        assert "/dev/dri/renderD128" in paths
        assert "/dev/dri/renderD131" in paths
        assert "Runtime" not in hc
        assert "DeviceRequests" not in hc
        assert hc["GroupAdd"]  # video/render groups
        assert created["Image"] == "rocm/dev"
        # env records the GPU set for inspect round-trips
        assert any(e.startswith("GDA_GPU_UUIDS=") for e in created["Env"])

        await rt.start("demo-1")
        st = await rt.inspect("demo-1")
        assert st.running and st.name == "demo-1"
        assert st.gpu_uuids == ["MockMI355X-0", "MockMI355X-3"]
        assert st.upper_dir == "/var/lib/docker/overlay2/x/diff"
        assert await rt.inspect("missing") is None

        out = await rt.execute("demo-1", ["echo", "x"])
        assert out == "hello-from-exec\n"

        vs = await rt.volume_create("vol-1", {"size": "10GB"})
        assert vs.mountpoint.endswith("vol-1/_data")

        # lifecycle verbs hit the right endpoints with the right params
        await rt.stop("demo-1", timeout=7)
        await rt.restart("demo-1", timeout=9)
        await rt.pause("demo-1")
        await rt.unpause("demo-1")
        await rt.remove("demo-1", force=True)
        await rt.commit("demo-1", "snap:v2")
        calls = [(m, p, q) for m, p, _b, q in fake.requests]
        assert ("POST", "/v1.41/containers/demo-1/stop", {"t": "7"}) in calls
        assert ("POST", "/v1.41/containers/demo-1/restart", {"t": "9"}) in calls
        assert ("POST", "/v1.41/containers/demo-1/pause", {}) in calls
        assert ("POST", "/v1.41/containers/demo-1/unpause", {}) in calls
        assert ("DELETE", "/v1.41/containers/demo-1", {"force": "true", "v": "false"}) in calls
        assert ("POST", "/v1.41/commit", {"container": "demo-1", "repo": "snap", "tag": "v2"}) in calls

        await rt.close()
        await runner.cleanup()

    run(main())
