set -x
exec > gpurun_out/callsoak.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cat > /tmp/fp8_only.py <<'PYEOF'
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
print(ext.gemm_fp8_mx_tflops(0, 4096, 6))
PYEOF
cd /tmp && export TMPDIR=/tmp
timeout 200 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_LDS SQ_LDS_BANK_CONFLICT -d $GRAFT_REPO_ROOT/gpurun_out/pmc_fp8v1 -o fp8v1 -- python /tmp/fp8_only.py 2>&1 | tail -1
cd $GRAFT_REPO_ROOT
timeout 330 python scripts/soak.py --seconds 300 > gpurun_out/soak_5min_r2.json 2>&1 || timeout 330 python scripts/soak.py > gpurun_out/soak_5min_r2.json 2>&1
tail -3 gpurun_out/soak_5min_r2.json
