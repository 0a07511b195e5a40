#!/usr/bin/env python3
"""MX-fp8 GEMM numerics race-screen + throughput on a real MI355X."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import torch

    from gpu_docker_api_amd.ops import hipcore

    ext = hipcore.load_ext()
    torch.manual_seed(9)
    out = {}
    for (M, N, K) in ((256, 256, 256), (512, 256, 512), (4096, 4096, 4096)):
        A = (torch.randn(M, K, device="cuda") * 0.5).to(torch.float8_e4m3fn)
        Bt = (torch.randn(N, K, device="cuda") * 0.5).to(torch.float8_e4m3fn)
        ref = A.float() @ Bt.float().T
        scale = ref.abs().max().item() + 1e-6
        for shape in (16, 32):
            worst = 0.0
            for _ in range(4):
                C = ext.gemm_fp8_mx(A.view(torch.uint8), Bt.view(torch.uint8), shape=shape)
                torch.cuda.synchronize()
                worst = max(worst, (C - ref).abs().max().item() / scale)
            out[f"relerr_s{shape}_{M}x{N}x{K}"] = worst
    for size, iters in ((4096, 8), (8192, 4)):
        # interleaved 16 vs 32 shape A/B
        for shape in (16, 32):
            out[f"tflops_s{shape}_{size}"] = []
        for _ in range(3):
            for shape in (16, 32):
                out[f"tflops_s{shape}_{size}"].append(
                    round(ext.gemm_fp8_mx_tflops(0, size, iters, shape=shape), 1))

    # fp4 (e2m1): pack random nibbles, dequant via LUT for the reference
    lut = torch.tensor(
        [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
         -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0], device="cuda")
    for (M, N, K) in ((256, 256, 512), (512, 256, 1024), (4096, 4096, 4096)):
        gen = torch.Generator(device="cuda").manual_seed(13)
        nibbles = torch.randint(0, 16, (M, K), generator=gen, device="cuda", dtype=torch.uint8)
        nibB = torch.randint(0, 16, (N, K), generator=gen, device="cuda", dtype=torch.uint8)
        packA = (nibbles[:, 0::2] | (nibbles[:, 1::2] << 4)).contiguous()
        packB = (nibB[:, 0::2] | (nibB[:, 1::2] << 4)).contiguous()
        ref = lut[nibbles.long()] @ lut[nibB.long()].T
        scale = ref.abs().max().item() + 1e-6
        for shape in (16, 32):
            worst = 0.0
            for _ in range(4):
                C = ext.gemm_fp4_mx(packA, packB, K, shape=shape)
                torch.cuda.synchronize()
                worst = max(worst, (C - ref).abs().max().item() / scale)
            out[f"fp4_relerr_s{shape}_{M}x{N}x{K}"] = worst
    for size, iters in ((4096, 8), (8192, 4)):
        for shape in (16, 32):
            out[f"fp4_tflops_s{shape}_{size}"] = []
        for _ in range(3):
            for shape in (16, 32):
                out[f"fp4_tflops_s{shape}_{size}"].append(
                    round(ext.gemm_fp4_mx_tflops(0, size, iters, shape=shape), 1))
    print(json.dumps(out))
    os.makedirs("gpurun_out", exist_ok=True)
    open("gpurun_out/fp8_mx.json", "w").write(json.dumps(out))


if __name__ == "__main__":
    main()
