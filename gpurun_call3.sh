set -x
exec > gpurun_out/call3.log 2>&1
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
rocprofv3 --list-avail 2>/dev/null | grep -oE "SQ_[A-Z_0-9]+" | sort -u > gpurun_out/sq_counters.txt
wc -l gpurun_out/sq_counters.txt
timeout 500 python scripts/gemm_ab.py 0,1,4,6
echo "=== PMC on v0 ==="
cat > /tmp/pmc_v0.py <<'PYEOF'
from gpu_docker_api_amd.ops import hipcore
ext = hipcore.load_ext()
print(ext.gemm_bf16_8ph_tflops(0, 4096, 4, 0))
PYEOF
cd /tmp && timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_INSTS_MFMA SQ_WAIT_INST_LDS SQ_WAIT_INST_VMEM SQ_INSTS_LDS SQ_INSTS_VALU -d $GRAFT_REPO_ROOT/gpurun_out/pmc_v0 -o v0 -- python /tmp/pmc_v0.py
echo "pmc rc=$?"
find $GRAFT_REPO_ROOT/gpurun_out/pmc_v0 -name "*.csv" | head
