set -x
exec > gpurun_out/call_prof.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cat > /tmp/mx_prof.py <<'PYEOF'
import sys
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
which = sys.argv[1]
if which == "fp8":
    print(ext.gemm_fp8_mx_tflops(0, 4096, 6))
else:
    print(ext.gemm_fp4_mx_tflops(0, 4096, 6))
PYEOF
cd /tmp && export TMPDIR=/tmp
timeout 200 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof_fp8 -o fp8 -- python /tmp/mx_prof.py fp8 2>&1 | grep -E "gemm|KERNEL|N/A" | head -5
timeout 200 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_LDS SQ_LDS_BANK_CONFLICT -d $GRAFT_REPO_ROOT/gpurun_out/pmc_fp8 -o fp8 -- python /tmp/mx_prof.py fp8 2>&1 | tail -1
timeout 200 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof_fp4 -o fp4 -- python /tmp/mx_prof.py fp4 2>&1 | grep -E "gemm|KERNEL|N/A" | head -5
timeout 200 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_LDS SQ_LDS_BANK_CONFLICT -d $GRAFT_REPO_ROOT/gpurun_out/pmc_fp4 -o fp4 -- python /tmp/mx_prof.py fp4 2>&1 | tail -1
cd $GRAFT_REPO_ROOT
echo "=== copybench ==="
timeout 200 python scripts/copybench.py > gpurun_out/copybench_r2.json 2>&1; echo rc=$?
tail -5 gpurun_out/copybench_r2.json
find gpurun_out/prof_fp8 gpurun_out/prof_fp4 -name "*stats*" 2>/dev/null | head
