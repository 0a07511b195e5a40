#!/usr/bin/env python3
"""Copy-engine benchmark: io_uring engine vs tar pipe vs shutil on a
synthetic rootfs-like tree (many small files + a few large + sparse).

This is the data path of rolling replacement (the reference shells out to
tar: /root/reference/utils/copy.go:17-27). Results go to BASELINE.md.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import shutil
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gpu_docker_api_amd.utils.copy import CopyEngine


def make_tree(root: str, small_files: int, large_mib: int) -> int:
    os.makedirs(root, exist_ok=True)
    total = 0
    blob4k = os.urandom(4096)
    for i in range(small_files):
        d = os.path.join(root, f"d{i % 16:02d}")
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, f"s{i:05d}.txt"), "wb") as f:
            f.write(blob4k)
        total += 4096
    blob1m = os.urandom(1024 * 1024)
    for i in range(large_mib):
        with open(os.path.join(root, f"large{i:03d}.bin"), "wb") as f:
            f.write(blob1m)
        total += len(blob1m)
    with open(os.path.join(root, "sparse.img"), "wb") as f:
        f.write(b"x")
        f.seek(64 * 1024 * 1024)
        f.write(b"y")
    return total


def drop_caches_hint():
    # no privileged cache drop; rely on fresh random content per run instead
    pass


async def bench(engine: str, src: str, dst_base: str, runs: int) -> dict:
    times = []
    eng = CopyEngine(engine)
    for r in range(runs):
        dst = os.path.join(dst_base, f"{engine}-{r}")
        shutil.rmtree(dst, ignore_errors=True)
        t0 = time.perf_counter()
        await eng.copy_dir(src, dst)
        times.append(time.perf_counter() - t0)
        shutil.rmtree(dst, ignore_errors=True)
    return {"engine": engine, "best_s": round(min(times), 3), "avg_s": round(sum(times) / len(times), 3)}


async def main_async(args):
    base = args.dir
    shutil.rmtree(base, ignore_errors=True)
    src = os.path.join(base, "src")
    total = make_tree(src, args.small_files, args.large_mib)
    out = {"tree_bytes": total, "small_files": args.small_files, "large_mib": args.large_mib, "results": []}
    for engine in ("iouring", "tar", "python"):
        try:
            res = await bench(engine, src, base, args.runs)
            res["gbps"] = round(total / res["best_s"] / 1e9, 3)
            out["results"].append(res)
        except Exception as exc:
            out["results"].append({"engine": engine, "error": str(exc)})
    print(json.dumps(out, indent=1))
    shutil.rmtree(base, ignore_errors=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--small-files", type=int, default=2000)
    p.add_argument("--large-mib", type=int, default=512)
    p.add_argument("--runs", type=int, default=3)
    p.add_argument("--dir", default="/tmp/gda-copybench")
    args = p.parse_args()
    asyncio.run(main_async(args))


if __name__ == "__main__":
    main()
