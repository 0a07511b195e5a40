set -x
exec > gpurun_out/call10.log 2>&1
export PYTHONPATH=$GRAFT_REPO_ROOT
cd $GRAFT_REPO_ROOT
timeout 480 python -m pytest tests -m gpu -q -rs -p no:cacheprovider 2>&1 | tail -4
echo "=== scenarios ==="
timeout 300 python scripts/scenarios.py > gpurun_out/scenarios_r2.json 2>&1; echo rc=$?
tail -20 gpurun_out/scenarios_r2.json
echo "=== loadtest ==="
timeout 300 python scripts/loadtest.py > gpurun_out/loadtest_r2.json 2>&1; echo rc=$?
tail -10 gpurun_out/loadtest_r2.json
