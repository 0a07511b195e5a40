#!/usr/bin/env python3
"""8-phase GEMM variant A/B + race screen on a real MI355X (guide §5.4
two-lane discipline: PIPE=1 is a new sync structure -> multi-run race screen
at 256/512/4096 + within-probe interleaved A/B vs the unmodified template)."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import torch

    from gpu_docker_api_amd.ops import hipcore

    variants = [int(x) for x in (sys.argv[1].split(",") if len(sys.argv) > 1 else "0,1,2,3".split(","))]
    ext = hipcore.load_ext()
    out = {"variants": variants, "race_screen": {}, "ab": {}}

    # race screen: every variant vs fp32 torch reference, repeated
    torch.manual_seed(11)
    for size in (256, 512, 4096):
        A = (torch.randn(size, size, device="cuda") * 0.5).bfloat16()
        Bt = (torch.randn(size, size, device="cuda") * 0.5).bfloat16()
        ref = A.float() @ Bt.float().T
        scale = ref.abs().max().item() + 1e-6
        for v in variants:
            worst = 0.0
            for rep in range(4):
                C = ext.gemm_bf16_8ph(A, Bt, variant=v)
                torch.cuda.synchronize()
                worst = max(worst, (C - ref).abs().max().item() / scale)
            out["race_screen"][f"v{v}_{size}"] = worst
            assert worst < 0.02, f"variant {v} size {size}: rel err {worst}"

    # interleaved A/B at both shapes
    for size, iters in ((4096, 4), (8192, 2)):
        r = ext.gemm_bf16_8ph_ab(device=0, size=size, iters=iters, rounds=3,
                                 variants=variants)
        out["ab"][size] = {int(k): [round(x, 1) for x in v] for k, v in r.items()}

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/gemm_ab.json", "w") as f:
        json.dump(out, f, indent=1)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
