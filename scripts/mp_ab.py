#!/usr/bin/env python3
"""Merged-phase MX variant A/B (shape 16 default vs 19/20 merged)."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import torch

    from gpu_docker_api_amd.ops import hipcore

    ext = hipcore.load_ext()
    out = {}
    torch.manual_seed(9)
    A = (torch.randn(512, 512, device="cuda") * 0.5).to(torch.float8_e4m3fn)
    Bt = (torch.randn(256, 512, device="cuda") * 0.5).to(torch.float8_e4m3fn)
    ref = A.float() @ Bt.float().T
    sc = ref.abs().max().item()
    for _ in range(4):  # race screen: new sync structure
        C = ext.gemm_fp8_mx(A.view(torch.uint8), Bt.view(torch.uint8), shape=20)
        torch.cuda.synchronize()
        err = (C - ref).abs().max().item() / sc
        assert err < 1e-3, err
    out["fp8_mp_relerr"] = err
    lut = torch.tensor(
        [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
         -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0], device="cuda")
    gen = torch.Generator(device="cuda").manual_seed(13)
    nA = torch.randint(0, 16, (512, 1024), generator=gen, device="cuda", dtype=torch.uint8)
    nB = torch.randint(0, 16, (256, 1024), generator=gen, device="cuda", dtype=torch.uint8)
    packA = (nA[:, 0::2] | (nA[:, 1::2] << 4)).contiguous()
    packB = (nB[:, 0::2] | (nB[:, 1::2] << 4)).contiguous()
    ref4 = lut[nA.long()] @ lut[nB.long()].T
    for _ in range(4):
        C = ext.gemm_fp4_mx(packA, packB, 1024, shape=19)
        torch.cuda.synchronize()
        assert torch.equal(C, ref4), "fp4 merged not exact"
    out["fp4_mp_exact"] = True

    for size, iters in ((4096, 6), (8192, 3)):
        for key in ("fp4_16", "fp4_19", "fp8_16", "fp8_20"):
            out.setdefault(f"{key}_{size}", [])
        for _ in range(3):
            # merged FIRST this run (position/clock-ramp bias control)
            out[f"fp4_19_{size}"].append(round(ext.gemm_fp4_mx_tflops(0, size, iters, shape=19), 1))
            out[f"fp4_16_{size}"].append(round(ext.gemm_fp4_mx_tflops(0, size, iters, shape=16), 1))
            out[f"fp8_20_{size}"].append(round(ext.gemm_fp8_mx_tflops(0, size, iters, shape=20), 1))
            out[f"fp8_16_{size}"].append(round(ext.gemm_fp8_mx_tflops(0, size, iters, shape=16), 1))
    print(json.dumps(out))
    os.makedirs("gpurun_out", exist_ok=True)
    open("gpurun_out/mp_ab.json", "w").write(json.dumps(out))


if __name__ == "__main__":
    main()
