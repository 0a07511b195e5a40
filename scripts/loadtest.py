#!/usr/bin/env python3
"""Concurrent-tenant load test against a live daemon: N asyncio tenants each
looping the create->patch->delete cycle for a fixed duration; reports global
p50/p95/p99 and throughput. Complements bench.py (which is the torchrun
contract) with a pure-concurrency view on one box."""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import shutil
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


async def tenant(client, name: str, gpu: int, stop_at: float, latencies: list):
    while time.perf_counter() < stop_at:
        t0 = time.perf_counter()
        r = await client.post(
            "/api/v1/replicaSet",
            json={
                "imageName": "synthetic:load",
                "replicaSetName": name,
                "gpuCount": gpu,
                "cpuCount": 1,
            },
        )
        assert r.json()["code"] == 200, r.text
        r = await client.patch(
            f"/api/v1/replicaSet/{name}", json={"memoryPatch": {"memory": "2GB"}}
        )
        assert r.json()["code"] == 200, r.text
        r = await client.delete(f"/api/v1/replicaSet/{name}")
        assert r.json()["code"] == 200, r.text
        latencies.append((time.perf_counter() - t0) * 1000.0)


async def main_async(args):
    import httpx

    from gpu_docker_api_amd.config import Config
    from gpu_docker_api_amd.routers.app import Daemon, build_app

    data_dir = "/tmp/gda-loadtest"
    shutil.rmtree(data_dir, ignore_errors=True)
    cfg = Config(
        state="memory",
        data_dir=data_dir,
        runtime="proc",
        inventory=args.inventory,
        mock_gpus=max(8, args.tenants),
        copy_engine="auto",
        port_range="46000-47000",
    )
    d = Daemon(cfg)
    await d.start()
    app = build_app(cfg, daemon=d)
    from gpu_docker_api_amd.routers.app import _mount

    _mount(app, d)

    transport = httpx.ASGITransport(app=app)
    client = httpx.AsyncClient(transport=transport, base_url="http://daemon")

    n_gpus = d.gpu.node_gpu_count
    stop_at = time.perf_counter() + args.seconds
    lat: list = []
    tasks = [
        tenant(client, f"ld{i}", 1 if i < n_gpus else 0, stop_at, lat)
        for i in range(args.tenants)
    ]
    t0 = time.perf_counter()
    await asyncio.gather(*tasks)
    wall = time.perf_counter() - t0
    lat.sort()
    out = {
        "tenants": args.tenants,
        "gpus": n_gpus,
        "cycles": len(lat),
        "wall_s": round(wall, 2),
        "cycles_per_s": round(len(lat) / wall, 1),
        "p50_ms": round(statistics.median(lat), 2),
        "p95_ms": round(lat[int(0.95 * len(lat))], 2),
        "p99_ms": round(lat[min(int(0.99 * len(lat)), len(lat) - 1)], 2),
    }
    print(json.dumps(out))
    await client.aclose()
    await d.stop()
    shutil.rmtree(data_dir, ignore_errors=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tenants", type=int, default=8)
    p.add_argument("--seconds", type=float, default=10.0)
    p.add_argument("--inventory", default="auto")
    args = p.parse_args()
    asyncio.run(main_async(args))


if __name__ == "__main__":
    main()
