"""End-to-end daemon process tests: the real `python -m gpu_docker_api_amd`
entry point, and the `--state etcd:<endpoint>` branch against a live (fake)
etcd gateway served over TCP."""
import json
import os
import socket
import subprocess
import sys
import threading
import time

import httpx
import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _wait_ping(port, timeout=60):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if httpx.get(f"http://127.0.0.1:{port}/ping", timeout=1).status_code == 200:
                return True
        except Exception:
            time.sleep(0.2)
    return False


def test_daemon_main_process(tmp_path):
    port = _free_port()
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "gpu_docker_api_amd",
            "--addr",
            f"127.0.0.1:{port}",
            "--runtime",
            "mock",
            "--inventory",
            "mock",
            "--dataDir",
            str(tmp_path / "data"),
            "--portRange",
            "49100-49200",
            "--logLevel",
            "warning",
        ],
        cwd=ROOT,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        start_new_session=True,
    )
    try:
        assert _wait_ping(port), "daemon did not come up"
        r = httpx.post(
            f"http://127.0.0.1:{port}/api/v1/replicaSet",
            json={"imageName": "img", "replicaSetName": "proc", "gpuCount": 2},
            timeout=30,
        ).json()
        assert r["code"] == 200 and r["data"]["name"] == "proc-1"
        gpus = httpx.get(f"http://127.0.0.1:{port}/api/v1/resources/gpus", timeout=10).json()
        assert sum(gpus["data"].values()) == 2
    finally:
        proc.terminate()  # exact pid
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=5)


def test_daemon_against_live_etcd_gateway(tmp_path, run):
    import uvicorn

    from gpu_docker_api_amd.config import Config
    from gpu_docker_api_amd.routers.app import Daemon
    from gpu_docker_api_amd.state.etcd_fake import build_fake_etcd

    etcd_port = _free_port()
    etcd_app = build_fake_etcd()
    server = uvicorn.Server(
        uvicorn.Config(etcd_app, host="127.0.0.1", port=etcd_port, log_level="error")
    )
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    deadline = time.time() + 30
    while time.time() < deadline and not server.started:
        time.sleep(0.05)

    async def main():
        from gpu_docker_api_amd.models import ContainerRun

        cfg = Config(
            state=f"etcd:http://127.0.0.1:{etcd_port}",
            data_dir=str(tmp_path / "d"),
            runtime="mock",
            inventory="mock",
            copy_engine="python",
            port_range="49300-49400",
        )
        d = Daemon(cfg)
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="etc", gpu_count=1)
        )
        info = await d.replicaset.get_container_info("etc")
        assert info["containerName"] == "etc-1"
        await d.queue.drain()
        await d.stop()
        # the spec and scheduler state really live in the remote etcd
        kvs = etcd_app.state.mvcc.range_prefix("/gpu-docker-api/apis/v1/")
        keys = [kv.key for kv in kvs]
        assert "/gpu-docker-api/apis/v1/containers/etc" in keys
        assert "/gpu-docker-api/apis/v1/gpus/gpuStatusMapKey" in keys

        # a second daemon boots from that etcd and sees the allocation
        d2 = Daemon(
            Config(
                state=f"etcd:http://127.0.0.1:{etcd_port}",
                data_dir=str(tmp_path / "d2"),
                runtime="mock",
                inventory="mock",
                port_range="49300-49400",
            )
        )
        await d2.start()
        assert sum(d2.gpu.get_gpu_status().values()) == 1
        assert d2.container_versions.get("etc") == 1
        await d2.stop()

    run(main())
    server.should_exit = True
    t.join(timeout=10)


def test_daemon_sigterm_persists_state_across_processes(tmp_path):
    """SIGTERM -> uvicorn lifespan shutdown -> Daemon.stop persists; a new
    daemon PROCESS on the same dataDir recovers allocations, versions and
    specs from the WAL (the reference's etcd-reload path, main.go:139-154,
    without the external etcd)."""

    def spawn(port):
        return subprocess.Popen(
            [sys.executable, "-m", "gpu_docker_api_amd",
             "--addr", f"127.0.0.1:{port}",
             "--runtime", "mock", "--inventory", "mock",
             "--dataDir", str(tmp_path / "data"),
             "--portRange", "49300-49400",
             "--logLevel", "warning"],
            cwd=ROOT,
            stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT,
            start_new_session=True,
        )

    port1 = _free_port()
    p1 = spawn(port1)
    p2 = None
    try:
        assert _wait_ping(port1), "daemon 1 did not come up"
        r = httpx.post(
            f"http://127.0.0.1:{port1}/api/v1/replicaSet",
            json={"imageName": "img", "replicaSetName": "dur", "gpuCount": 3},
            timeout=30,
        ).json()
        assert r["code"] == 200
        p1.terminate()  # SIGTERM: graceful lifespan shutdown
        assert p1.wait(timeout=20) is not None

        port2 = _free_port()
        p2 = spawn(port2)
        assert _wait_ping(port2), "daemon 2 did not come up"
        gpus = httpx.get(
            f"http://127.0.0.1:{port2}/api/v1/resources/gpus", timeout=10
        ).json()
        assert sum(gpus["data"].values()) == 3  # allocation survived
        info = httpx.get(
            f"http://127.0.0.1:{port2}/api/v1/replicaSet/dur", timeout=10
        ).json()
        assert info["code"] == 200 and info["data"]["version"] == 1
        hist = httpx.get(
            f"http://127.0.0.1:{port2}/api/v1/replicaSet/dur/history", timeout=10
        ).json()
        assert [h["version"] for h in hist["data"]] == [1]
    finally:
        for p in (p1, p2):
            if p is not None:
                try:
                    p.terminate()
                    p.wait(timeout=10)
                except Exception:
                    try:
                        p.kill()
                        p.wait(timeout=5)
                    except Exception:
                        pass


def test_config_file_layering(tmp_path):
    """YAML config (the reference documents etc/config.yaml but never
    implemented it) with flag precedence: defaults < file < flags."""
    from gpu_docker_api_amd.__main__ import parse_args

    cfgf = tmp_path / "config.yaml"
    cfgf.write_text(
        "addr: 127.0.0.1:9999\n"
        "runtime: proc\n"
        "portRange: 50000-50100\n"      # camelCase accepted
        "mock_gpus: 4\n"                # snake_case accepted
        "xgmiProbe: true\n"
        "apikey: sekrit\n"
    )
    c = parse_args(["--config", str(cfgf)])
    assert c.addr == "127.0.0.1:9999"
    assert c.runtime == "proc"
    assert c.port_range == "50000-50100"
    assert c.mock_gpus == 4
    assert c.run_xgmi_probe is True
    assert c.apikey == "sekrit"
    # explicit flag beats the file
    c = parse_args(["--config", str(cfgf), "--runtime", "mock", "--addr", "0.0.0.0:1"])
    assert c.runtime == "mock" and c.addr == "0.0.0.0:1"
    assert c.port_range == "50000-50100"  # file still layers under
    # unknown keys are an error, not a silent ignore
    bad = tmp_path / "bad.yaml"
    bad.write_text("runtme: proc\n")
    import pytest as _pytest

    with _pytest.raises(SystemExit):
        parse_args(["--config", str(bad)])
