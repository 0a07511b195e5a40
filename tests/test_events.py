"""Live state-change event stream (SSE extension)."""
import asyncio
import json

from gpu_docker_api_amd.models import ContainerRun
from gpu_docker_api_amd.routers.app import Daemon
from helpers import make_config


def test_subscribe_receives_mutations(tmp_path, run):
    async def main():
        d = Daemon(make_config(tmp_path))
        await d.start()
        q, unsubscribe = d.store.subscribe()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="ev", gpu_count=1)
        )
        await d.queue.drain()
        events = []
        try:
            while True:
                events.append(await asyncio.wait_for(q.get(), timeout=2.0))
        except asyncio.TimeoutError:
            pass
        kinds = {(e["resource"], e["name"], e["type"]) for e in events}
        assert ("containers", "ev", "put") in kinds
        assert ("gpus", "gpuStatusMapKey", "put") in kinds
        assert ("versions", "containerVersionMapKey", "put") in kinds
        await d.replicaset.delete_container("ev")
        found_delete = False
        try:
            while True:
                e = await asyncio.wait_for(q.get(), timeout=2.0)
                if e["resource"] == "containers" and e["type"] == "delete":
                    found_delete = True
        except asyncio.TimeoutError:
            pass
        assert found_delete
        unsubscribe()
        await d.stop()

    run(main())


def test_sse_route_streams_over_tcp(tmp_path):
    """Real uvicorn server: httpx's ASGI transport buffers streaming bodies,
    so SSE must be tested over TCP."""
    import socket
    import threading
    import time

    import httpx
    import uvicorn

    from gpu_docker_api_amd.routers.app import build_app

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()

    app = build_app(make_config(tmp_path))
    server = uvicorn.Server(
        uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error")
    )
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    base = f"http://127.0.0.1:{port}"
    deadline = time.time() + 30
    while time.time() < deadline:
        try:
            if httpx.get(base + "/ping", timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.1)

    got = []
    try:
        with httpx.stream("GET", base + "/api/v1/events", timeout=20) as resp:
            assert resp.status_code == 200
            assert resp.headers["content-type"].startswith("text/event-stream")
            httpx.post(
                base + "/api/v1/replicaSet",
                json={"imageName": "img", "replicaSetName": "sse"},
                timeout=20,
            )
            for line in resp.iter_lines():
                if line.startswith("data: "):
                    got.append(json.loads(line[6:]))
                    if any(
                        e["resource"] == "containers" and e["name"] == "sse"
                        for e in got
                    ):
                        break
    finally:
        server.should_exit = True
        t.join(timeout=10)
    assert any(e["name"] == "sse" and e["type"] == "put" for e in got)
