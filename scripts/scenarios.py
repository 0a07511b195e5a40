#!/usr/bin/env python3
"""Measure the five BASELINE.json configs end-to-end and print one JSON
summary. Runs anywhere (mock inventory on CPU boxes, amdsmi on MI355X);
results are recorded in BASELINE.md.

  1. 0-GPU replicaSet run + volume create (plumbing, no GPU)
  2. 1-GPU run, then patch to 2 GPUs (rolling replace)
  3. volume 10GiB -> 50GiB resize with data copy; replicaSet picks it up
  4. 8 concurrent 1-GPU replicaSets saturate the node; Resource reports 0 free
  5. 4-GPU replicaSet commit + rollback across 3 versions with migration
"""
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

from gpu_docker_api_amd.config import Config
from gpu_docker_api_amd.models import (
    Bind,
    ContainerCommit,
    ContainerRun,
    GpuPatch,
    PatchRequest,
    VolumeCreate,
)
from gpu_docker_api_amd.routers.app import Daemon


async def timed(coro):
    t0 = time.perf_counter()
    result = await coro
    return (time.perf_counter() - t0) * 1000.0, result


async def fresh_daemon(base: str, tag: str, gpus_needed: int = 0) -> Daemon:
    data_dir = os.path.join(base, tag)
    shutil.rmtree(data_dir, ignore_errors=True)
    import torch

    use_gpu = torch.cuda.is_available() and gpus_needed <= torch.cuda.device_count()
    cfg = Config(
        state="memory",
        data_dir=data_dir,
        runtime="proc",
        inventory="amdsmi" if use_gpu else "mock",
        mock_gpus=8,
        copy_engine="auto",
        port_range="44000-45000",
    )
    d = Daemon(cfg)
    await d.start()
    return d


def fill_dir(path: str, mib: int) -> None:
    os.makedirs(path, exist_ok=True)
    blob = os.urandom(1024 * 1024)
    for i in range(mib):
        with open(os.path.join(path, f"f{i:04d}.bin"), "wb") as f:
            f.write(blob)


async def scenario1(base) -> dict:
    d = await fresh_daemon(base, "s1")
    reps = []
    for i in range(10):
        ms_run, _ = await timed(
            d.replicaset.run_gpu_container(
                ContainerRun(image_name="img", replica_set_name=f"plain{i}")
            )
        )
        ms_vol, _ = await timed(d.volume.create_volume(VolumeCreate(name=f"v{i}", size="1GB")))
        reps.append(ms_run + ms_vol)
        await d.replicaset.delete_container(f"plain{i}")
        await d.volume.delete_volume(f"v{i}")
    await d.stop()
    return {"p50_ms": round(statistics.median(reps), 2), "iters": len(reps)}


async def scenario2(base) -> dict:
    d = await fresh_daemon(base, "s2", gpus_needed=2)
    n_gpus = d.gpu.node_gpu_count
    target = 2 if n_gpus >= 2 else 0
    runs, patches = [], []
    for i in range(10):
        ms_run, _ = await timed(
            d.replicaset.run_gpu_container(
                ContainerRun(image_name="img", replica_set_name="job", gpu_count=1)
            )
        )
        ms_patch, _ = await timed(
            d.replicaset.patch_container("job", PatchRequest(gpu_patch=GpuPatch(gpu_count=target)))
        )
        runs.append(ms_run)
        patches.append(ms_patch)
        await d.replicaset.delete_container("job")
    await d.stop()
    return {
        "p50_run_ms": round(statistics.median(runs), 2),
        "p50_patch_ms": round(statistics.median(patches), 2),
        "patch_target_gpus": target,
    }


async def scenario3(base, payload_mib: int) -> dict:
    d = await fresh_daemon(base, "s3")
    await d.volume.create_volume(VolumeCreate(name="data", size="10GB"))
    vs = await d.runtime.volume_inspect("data-1")
    fill_dir(vs.mountpoint, payload_mib)
    await d.replicaset.run_gpu_container(
        ContainerRun(
            image_name="img",
            replica_set_name="consumer",
            binds=[Bind(src="data-1", dest="/data")],
        )
    )
    ms_resize, out = await timed(d.volume.patch_volume_size("data", "50GB"))
    # replicaSet picks up the new volume
    ms_rebind, _ = await timed(
        d.replicaset.patch_container(
            "consumer",
            PatchRequest(
                volume_patch={
                    "oldBind": {"src": "data-1", "dest": "/data"},
                    "newBind": {"src": out["name"], "dest": "/data"},
                }
            ),
        )
    )
    ok = (await d.runtime.inspect("consumer-2")).binds == [f"{out['name']}:/data"]
    await d.stop()
    return {
        "payload_mib": payload_mib,
        "resize_ms": round(ms_resize, 2),
        "rebind_patch_ms": round(ms_rebind, 2),
        "bind_updated": ok,
    }


async def scenario4(base) -> dict:
    d = await fresh_daemon(base, "s4")
    n = d.gpu.node_gpu_count
    t0 = time.perf_counter()
    await asyncio.gather(
        *[
            d.replicaset.run_gpu_container(
                ContainerRun(image_name="img", replica_set_name=f"w{i}", gpu_count=1)
            )
            for i in range(n)
        ]
    )
    wall_ms = (time.perf_counter() - t0) * 1000.0
    free = sum(1 for v in d.gpu.get_gpu_status().values() if v == 0)
    denied = False
    try:
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="extra", gpu_count=1)
        )
    except Exception:
        denied = True
    await d.stop()
    return {"n_concurrent": n, "wall_ms": round(wall_ms, 2), "free_after": free, "overflow_denied": denied}


async def scenario5(base, payload_mib: int) -> dict:
    d = await fresh_daemon(base, "s5", gpus_needed=4)
    n_gpus = min(4, d.gpu.node_gpu_count)
    await d.replicaset.run_gpu_container(
        ContainerRun(image_name="img", replica_set_name="big", gpu_count=n_gpus)
    )
    st = await d.runtime.inspect("big-1")
    fill_dir(os.path.join(st.upper_dir, "work"), payload_mib)
    ms_commit, _ = await timed(
        d.replicaset.commit_container("big", ContainerCommit(new_image_name="big-snap"))
    )
    # build 3 versions, then roll back across them
    await d.replicaset.patch_container("big", PatchRequest(gpu_patch=GpuPatch(gpu_count=max(n_gpus - 1, 0))))
    await d.replicaset.patch_container("big", PatchRequest(gpu_patch=GpuPatch(gpu_count=max(n_gpus - 2, 0))))
    ms_rollback, out = await timed(d.replicaset.rollback_container("big", 1))
    st = await d.runtime.inspect(out["containerName"])
    data_ok = os.path.exists(os.path.join(st.upper_dir, "work", "f0000.bin"))
    await d.stop()
    return {
        "gpus": n_gpus,
        "payload_mib": payload_mib,
        "commit_ms": round(ms_commit, 2),
        "rollback_ms": round(ms_rollback, 2),
        "versions": 4,
        "data_migrated": data_ok,
    }


async def main_async(args):
    base = args.data_dir
    shutil.rmtree(base, ignore_errors=True)
    out = {
        "scenario1_plumbing": await scenario1(base),
        "scenario2_gpu_rescale": await scenario2(base),
        "scenario3_volume_resize": await scenario3(base, args.payload_mib),
        "scenario4_saturation": await scenario4(base),
        "scenario5_commit_rollback": await scenario5(base, args.payload_mib),
    }
    print(json.dumps(out, indent=1))
    shutil.rmtree(base, ignore_errors=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--payload-mib", type=int, default=64, help="data-disk payload for scenarios 3/5")
    p.add_argument("--data-dir", default="/tmp/gda-scenarios")
    args = p.parse_args()
    asyncio.run(main_async(args))


if __name__ == "__main__":
    main()
