set -x
exec > gpurun_out/call5.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cd $GRAFT_REPO_ROOT
timeout 600 python scripts/gemm_ab.py 0,8,10
cat > /tmp/zf.py <<'PYEOF'
import json, torch
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
out = {}
for name, fill in (("zero", torch.zeros), ("rand", None)):
    for size in (4096, 8192):
        if fill is None:
            torch.manual_seed(3)
            A = (torch.randn(size, size, device="cuda") * 0.5).bfloat16()
            Bt = (torch.randn(size, size, device="cuda") * 0.5).bfloat16()
        else:
            A = fill(size, size, device="cuda", dtype=torch.bfloat16)
            Bt = fill(size, size, device="cuda", dtype=torch.bfloat16)
        for _ in range(3):
            ext.gemm_bf16_8ph(A, Bt, variant=0)
        torch.cuda.synchronize()
        t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        n = 8 if size == 4096 else 3
        for _ in range(n):
            C = ext.gemm_bf16_8ph(A, Bt, variant=0)
        t1.record(); torch.cuda.synchronize()
        ms = t0.elapsed_time(t1) / n
        out[f"{name}_{size}"] = round(2.0 * size**3 / (ms * 1e9), 1)
print(json.dumps(out))
open("gpurun_out/zerofill.json", "w").write(json.dumps(out))
PYEOF
( for i in $(seq 1 12); do sleep 4; rocm-smi --showgpuclocks --showpower 2>/dev/null | grep -E "sclk|Power" ; done > gpurun_out/clocks_during.txt ) &
CLK=$!
timeout 300 python /tmp/zf.py
wait $CLK
