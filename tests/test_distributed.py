"""Multi-process distributed path on CPU (gloo, world_size 2): the bench
driver's torchrun topology — rank 0 serves the daemon, all ranks are
concurrent tenants."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_two_ranks_gloo():
    env = dict(os.environ)
    env.pop("ROCR_VISIBLE_DEVICES", None)
    out = subprocess.run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node",
            "2",
            "--master-addr",
            "127.0.0.1",
            "--master-port",
            "29613",
            os.path.join(ROOT, "bench.py"),
            "--gpus",
            "2",
            "--steps",
            "3",
            "--warmup",
            "1",
        ],
        capture_output=True,
        text=True,
        timeout=600,
        cwd=ROOT,
        env=env,
    )
    assert out.returncode == 0, (out.stdout + out.stderr)[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, out.stdout[-2000:]
    result = json.loads(lines[-1])
    assert result["n_gpus"] == 2
    assert result["config"]["global_batch"] == 2
    assert result["value"] > 0
    # 2 ranks x 3 steps, all cycles merged into the percentile set
    assert result["steps"] == 3
