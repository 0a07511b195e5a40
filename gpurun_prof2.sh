set -x
exec > gpurun_out/prof2.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cat > /tmp/shipped.py <<'PYEOF'
import sys
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
w = sys.argv[1]
if w == "fp8": print(ext.gemm_fp8_mx_tflops(0, 4096, 6))
elif w == "fp4": print(ext.gemm_fp4_mx_tflops(0, 4096, 6))
else: print(ext.gemm_bf16_8ph_tflops(0, 4096, 6))
PYEOF
cd /tmp && export TMPDIR=/tmp
for k in bf16 fp8 fp4; do
  timeout 200 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_LDS SQ_LDS_BANK_CONFLICT -d $GRAFT_REPO_ROOT/gpurun_out/pmc_ship_$k -o $k -- python /tmp/shipped.py $k 2>&1 | grep -E "^[0-9]" | tail -1
done
