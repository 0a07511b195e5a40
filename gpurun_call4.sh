set -x
exec > gpurun_out/call4.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cat > /tmp/pmc_v0.py <<'PYEOF'
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
print("tflops:", ext.gemm_bf16_8ph_tflops(0, 4096, 6, 0))
PYEOF
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_LDS SQ_ACTIVE_INST_LDS SQ_ACTIVE_INST_VMEM -d $GRAFT_REPO_ROOT/gpurun_out/pmc_v0 -o v0 -- python /tmp/pmc_v0.py
echo "pmc rc=$?"
for f in $(find $GRAFT_REPO_ROOT/gpurun_out/pmc_v0 -name "*.csv"); do echo "== $f"; head -3 "$f"; done
cd $GRAFT_REPO_ROOT
cat > /tmp/warm.py <<'PYEOF'
import json
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
out = {}
# clock-ramp study: consecutive short probes, then one long sustained probe
out["ramp_4096"] = [round(ext.gemm_bf16_8ph_tflops(0, 4096, 4, 0), 1) for _ in range(6)]
out["sustained_4096_it40"] = round(ext.gemm_bf16_8ph_tflops(0, 4096, 40, 0), 1)
out["sustained_8192_it10"] = round(ext.gemm_bf16_8ph_tflops(0, 8192, 10, 0), 1)
print(json.dumps(out))
open("gpurun_out/warm.json", "w").write(json.dumps(out))
PYEOF
timeout 400 python /tmp/warm.py
