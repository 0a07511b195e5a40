"""Python facade over the _hipcore extension + the rccl_smoke binary.

Fails loudly when a GPU is present but the native extension is missing
(NativeOpUnavailable) — GPU paths must never silently fall back.
"""
from __future__ import annotations

import asyncio
import json
import os
import subprocess
from typing import Optional

from ..xerrors import NativeOpUnavailable

_ext = None
_ext_err: Optional[str] = None


def load_ext():
    """Load the _hipcore torch extension (torch must be imported first so
    libtorch*.so are resolvable)."""
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    if _ext_err is not None:
        raise NativeOpUnavailable(_ext_err)
    try:
        import importlib.util
        import torch  # noqa: F401  (loads libtorch into the process)

        so = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_hipcore.so")
        if not os.path.exists(so):
            raise FileNotFoundError(f"{so} not built (run ops.build)")
        spec = importlib.util.spec_from_file_location("_hipcore", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
        return _ext
    except Exception as exc:
        _ext_err = f"native _hipcore unavailable: {exc}"
        raise NativeOpUnavailable(_ext_err) from exc


def gpu_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def run_probe(mib: int = 256, iters: int = 5) -> dict:
    """Measure HBM stream + pairwise p2p bandwidth across visible GPUs.

    Returns {"gpus": [uuid...], "hbm_gbps": [...], "p2p_gbps": [[...]]} —
    the shape parallel.topology.Topology.overlay_measured consumes. GPU
    identity comes from torch (HIP enumeration order); UUIDs are taken from
    the amdsmi inventory at matching indices when available.
    """
    ext = load_ext()
    n = ext.device_count()
    if n == 0:
        raise NativeOpUnavailable("no HIP devices visible")
    uuids = [f"GPU-{i}" for i in range(n)]
    try:
        from ..parallel.inventory import AmdSmiInventory

        infos = AmdSmiInventory().enumerate()
        if len(infos) == n:
            uuids = [g.uuid for g in infos]
    except Exception:
        pass
    # one-shot matrix: per-device buffers allocated once (on an 8-GPU node
    # per-pair re-allocation would cost minutes of daemon startup)
    mat = ext.p2p_matrix(mib, iters)
    hbm = [round(mat[i][i], 1) for i in range(n)]
    p2p = [
        [0.0 if i == j else round(mat[i][j], 1) for j in range(n)] for i in range(n)
    ]
    return {"gpus": uuids, "hbm_gbps": hbm, "p2p_gbps": p2p}


async def run_probe_async(mib: int = 256, iters: int = 5) -> dict:
    return await asyncio.get_running_loop().run_in_executor(
        None, lambda: run_probe(mib, iters)
    )


def validate_gpus(
    indices: Optional[list] = None, size: int = 4096, iters: int = 5, fp8: bool = True
) -> dict:
    """Per-GPU health burn-in: HBM stream bandwidth + dense bf16 MFMA GEMM
    (the 8-phase 256^2 structure, ~1.1 PF) + the MX-fp8 path (block-scaled
    mfma_scale 16x16x128, ~1.9 PF — exercised because fp8 training is what
    tenants run on MI355X and its datapath can fail independently of bf16).
    Run on FREE GPUs before scheduling; a GPU far below its siblings is
    flagged."""
    ext = load_ext()
    n = ext.device_count()
    idx = list(indices) if indices else list(range(n))
    size = max(256, (size // 256) * 256)  # kernel tile constraints
    report = {"gpus": []}
    for i in idx:
        if i < 0 or i >= n:
            continue
        hbm = ext.stream_bandwidth_gbps(i, 1024, 5)
        # prefer the 8-phase 256^2 pipelined structure (~1000 TF) when the
        # shape allows; fall back to the 128^2 double-buffered kernel
        if size % 256 == 0 and size >= 192:
            tflops = ext.gemm_bf16_8ph_tflops(i, size, iters)
        else:
            tflops = ext.gemm_bf16_tflops(i, size, iters)
        entry = {
            "index": i,
            "hbm_gbps": round(hbm, 1),
            "bf16_tflops": round(tflops, 1),
        }
        if fp8 and size % 256 == 0 and size >= 256:
            entry["fp8_tflops"] = round(ext.gemm_fp8_mx_tflops(i, size, iters), 1)
        if fp8 and size % 256 == 0 and size >= 512:
            entry["fp4_tflops"] = round(ext.gemm_fp4_mx_tflops(i, size, iters), 1)
        report["gpus"].append(entry)
    vals = [g["bf16_tflops"] for g in report["gpus"]]
    if vals:
        top = max(vals)
        for g in report["gpus"]:
            g["healthy"] = bool(g["bf16_tflops"] > 0.7 * top and g["hbm_gbps"] > 2000)
    return report


async def validate_gpus_async(indices: Optional[list] = None, size: int = 4096, iters: int = 5) -> dict:
    return await asyncio.get_running_loop().run_in_executor(
        None, lambda: validate_gpus(indices, size, iters)
    )


def rccl_smoke_path() -> str:
    root = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    return os.path.join(root, "csrc", "bin", "rccl_smoke")


def run_rccl_smoke(ndev: int = 0, mib: int = 64, timeout: float = 120.0) -> dict:
    """Run the native RCCL all-reduce smoke binary; returns its JSON verdict."""
    binary = rccl_smoke_path()
    if not os.path.exists(binary):
        raise NativeOpUnavailable(f"{binary} not built (run ops.build)")
    cmd = [binary]
    if ndev > 0:
        cmd.append(str(ndev))
        cmd.append(str(mib))
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)
    line = (out.stdout.strip().splitlines() or ["{}"])[-1]
    try:
        result = json.loads(line)
    except json.JSONDecodeError:
        result = {"ok": False, "error": f"unparseable output: {line!r}", "rc": out.returncode}
    result.setdefault("ok", False)
    return result


async def run_rccl_smoke_async(ndev: int = 0, mib: int = 64) -> dict:
    return await asyncio.get_running_loop().run_in_executor(
        None, lambda: run_rccl_smoke(ndev, mib)
    )
