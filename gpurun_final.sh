set -x
exec > gpurun_out/final.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cd $GRAFT_REPO_ROOT
timeout 420 python -m pytest tests -m gpu -q -p no:cacheprovider 2>&1 | tail -1
timeout 180 python bench.py --gpus 1 --steps 64 --warmup 8 > gpurun_out/bench_final_proc.json 2>/dev/null; echo proc_rc=$?
timeout 180 python bench.py --gpus 1 --steps 48 --warmup 6 --runtime docker --port 18801 > gpurun_out/bench_final_docker.json 2>/dev/null; echo docker_rc=$?
timeout 300 python -c "
from gpu_docker_api_amd.ops import hipcore
import json
r = hipcore.validate_gpus(size=4096, iters=5)
print(json.dumps(r))
open('gpurun_out/validate_final.json','w').write(json.dumps(r))
"
tail -c 300 gpurun_out/bench_final_proc.json; echo; tail -c 200 gpurun_out/bench_final_docker.json
