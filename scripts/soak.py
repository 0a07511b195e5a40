#!/usr/bin/env python3
"""Stability soak: sustained multi-tenant load + chaos (random container
kills; the proc-runtime supervisor must resurrect unless-stopped workloads).
Reports latency drift and resource-leak indicators (fds, daemon RSS)."""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import shutil
import signal
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def fd_count() -> int:
    try:
        return len(os.listdir("/proc/self/fd"))
    except OSError:
        return -1


def rss_mb() -> float:
    with open("/proc/self/status") as f:
        for line in f:
            if line.startswith("VmRSS"):
                return int(line.split()[1]) / 1024.0
    return -1.0


async def main_async(args):
    import httpx

    from gpu_docker_api_amd.config import Config
    from gpu_docker_api_amd.routers.app import Daemon, build_app, _mount

    data_dir = "/tmp/gda-soak"
    shutil.rmtree(data_dir, ignore_errors=True)
    cfg = Config(
        state="memory",
        data_dir=data_dir,
        runtime="proc",
        inventory="auto",
        copy_engine="auto",
        port_range="48000-49000",
    )
    d = Daemon(cfg)
    await d.start()
    app = build_app(cfg, daemon=d)
    _mount(app, d)
    client = httpx.AsyncClient(transport=httpx.ASGITransport(app=app), base_url="http://s")

    n_gpus = d.gpu.node_gpu_count
    stop_at = time.perf_counter() + args.seconds
    windows: list[list[float]] = [[]]
    window_end = time.perf_counter() + args.seconds / 4
    kills = 0
    resurrections = 0

    # one long-lived replicaSet for chaos: the supervisor must resurrect it
    r = await client.post(
        "/api/v1/replicaSet",
        json={"imageName": "synthetic:victim", "replicaSetName": "victim", "gpuCount": min(1, n_gpus)},
    )
    assert r.json()["code"] == 200, r.text

    async def tenant(i: int):
        nonlocal window_end
        name = f"soak{i}"
        while time.perf_counter() < stop_at:
            t0 = time.perf_counter()
            r = await client.post(
                "/api/v1/replicaSet",
                json={"imageName": "synthetic:soak", "replicaSetName": name, "cpuCount": 1},
            )
            assert r.json()["code"] == 200, r.text
            r = await client.patch(
                f"/api/v1/replicaSet/{name}", json={"memoryPatch": {"memory": "2GB"}}
            )
            assert r.json()["code"] == 200, r.text
            r = await client.delete(f"/api/v1/replicaSet/{name}")
            assert r.json()["code"] == 200, r.text
            now = time.perf_counter()
            if now > window_end:
                windows.append([])
                window_end = now + args.seconds / 4
            windows[-1].append((now - t0) * 1000.0)

    async def chaos():
        nonlocal kills, resurrections
        while time.perf_counter() < stop_at:
            await asyncio.sleep(args.seconds / 8)
            st = await d.runtime.inspect("victim-1")
            if st and st.running and st.pid:
                try:
                    os.kill(st.pid, signal.SIGKILL)  # exact pid, never a pattern
                    kills += 1
                except ProcessLookupError:
                    pass
                for _ in range(60):
                    await asyncio.sleep(0.1)
                    st = await d.runtime.inspect("victim-1")
                    if st and st.running:
                        resurrections += 1
                        break

    await asyncio.gather(*(tenant(i) for i in range(args.tenants)), chaos())

    per_window = [round(statistics.median(w), 2) for w in windows if w]
    victim = await d.runtime.inspect("victim-1")
    out = {
        "seconds": args.seconds,
        "tenants": args.tenants,
        "cycles": sum(len(w) for w in windows),
        "p50_per_quarter_ms": per_window,
        "latency_drift": bool(per_window and per_window[-1] > per_window[0] * 1.5),
        "chaos_kills": kills,
        "supervisor_resurrections": resurrections,
        "victim_running_at_end": bool(victim and victim.running),
        "fds": fd_count(),
        "rss_mb": round(rss_mb(), 1),
    }
    print(json.dumps(out))
    await client.aclose()
    await d.stop()
    shutil.rmtree(data_dir, ignore_errors=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tenants", type=int, default=4)
    p.add_argument("--seconds", type=float, default=60.0)
    args = p.parse_args()
    asyncio.run(main_async(args))


if __name__ == "__main__":
    main()
