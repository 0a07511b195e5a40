set -x
exec > gpurun_out/call7.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cd $GRAFT_REPO_ROOT
timeout 420 python -m pytest tests -m gpu -q -rs -p no:cacheprovider 2>&1 | tail -8
echo "=== bench proc ==="
timeout 300 python bench.py --gpus 1 --steps 48 --warmup 8 > gpurun_out/bench_proc.json 2> gpurun_out/bench_proc.err
echo rc=$?
echo "=== bench docker(sim) ==="
timeout 300 python bench.py --gpus 1 --steps 48 --warmup 8 --runtime docker --port 18799 > gpurun_out/bench_docker.json 2> gpurun_out/bench_docker.err
echo rc=$?
tail -c 600 gpurun_out/bench_proc.json; echo; tail -c 600 gpurun_out/bench_docker.json
