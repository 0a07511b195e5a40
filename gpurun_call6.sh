set -x
exec > gpurun_out/call6.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cd $GRAFT_REPO_ROOT
cat > /tmp/sustained.py <<'PYEOF'
import json
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
out = {}
out["sustained_8192_it40_v0"] = round(ext.gemm_bf16_8ph_tflops(0, 8192, 40, 0), 1)
out["sustained_4096_it40_v0"] = round(ext.gemm_bf16_8ph_tflops(0, 4096, 40, 0), 1)
print(json.dumps(out))
open("gpurun_out/sustained.json", "w").write(json.dumps(out))
PYEOF
( for i in $(seq 1 120); do
    sleep 0.5
    cat /sys/class/drm/card*/device/pp_dpm_sclk 2>/dev/null | grep '\*'
    rocm-smi --showpower 2>/dev/null | grep -oE "Power \(W\): [0-9.]+"
  done > gpurun_out/clocks_fast.txt 2>&1 ) &
CLK=$!
timeout 300 python /tmp/sustained.py
kill $CLK 2>/dev/null
sort gpurun_out/clocks_fast.txt | uniq -c | sort -rn | head -12
