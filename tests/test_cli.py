"""CLI client against a live daemon over a real TCP socket."""
import json
import os
import shutil
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


@pytest.fixture
def live_daemon(tmp_path):
    import uvicorn

    from gpu_docker_api_amd.routers.app import build_app
    from helpers import make_config

    port = _free_port()
    cfg = make_config(tmp_path)
    app = build_app(cfg)
    server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error"))
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    deadline = time.time() + 30
    while time.time() < deadline:
        try:
            if httpx.get(f"http://127.0.0.1:{port}/ping", timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.1)
    yield port
    server.should_exit = True
    t.join(timeout=10)


def _cli(port, *args):
    env = dict(os.environ, GDA_ADDR=f"http://127.0.0.1:{port}")
    return subprocess.run(
        [sys.executable, "-m", "gpu_docker_api_amd.cli", *args],
        capture_output=True,
        text=True,
        timeout=60,
        cwd=ROOT,
        env=env,
    )


def test_cli_full_lifecycle(live_daemon):
    port = live_daemon
    out = _cli(port, "run", "job", "--image", "img:1", "--gpus", "1", "--memory", "1GB")
    assert out.returncode == 0, out.stderr
    assert json.loads(out.stdout)["name"] == "job-1"

    out = _cli(port, "ps")
    rows = json.loads(out.stdout)
    assert rows[0]["name"] == "job" and rows[0]["gpuCount"] == 1

    out = _cli(port, "stats", "job")
    st = json.loads(out.stdout)
    assert st["running"] is True and len(st["gpus"]) == 1

    out = _cli(port, "logs", "job", "--tail", "5")
    assert out.returncode == 0  # mock runtime: empty log, command succeeds

    out = _cli(port, "patch", "job", "--gpus", "2")
    assert json.loads(out.stdout)["containerName"] == "job-2"

    out = _cli(port, "history", "job")
    assert [h["version"] for h in json.loads(out.stdout)] == [2, 1]

    out = _cli(port, "resources", "gpus")
    assert sum(json.loads(out.stdout).values()) == 2

    out = _cli(port, "volume", "create", "vol", "--size", "5GB")
    assert json.loads(out.stdout)["name"] == "vol-1"

    out = _cli(port, "volume", "ls")
    vols = json.loads(out.stdout)
    assert vols[0]["name"] == "vol" and vols[0]["size"] == "5GB"

    out = _cli(port, "delete", "job")
    assert out.returncode == 0

    # error path: duplicate volume -> nonzero exit, message on stderr
    out = _cli(port, "volume", "create", "vol", "--size", "5GB")
    assert out.returncode == 1
    assert "1103" in out.stderr
