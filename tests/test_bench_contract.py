"""bench.py single-rank contract: one JSON line with the driver's schema."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "4", "--warmup", "1"],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=ROOT,
    )
    assert out.returncode == 0, (out.stdout + out.stderr)[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got {len(lines)}"
    r = json.loads(lines[0])
    for key in (
        "metric",
        "value",
        "unit",
        "n_gpus",
        "steps",
        "warmup",
        "ms_per_step",
        "higher_is_better",
        "scaling",
        "vs_baseline",
        "dtype",
        "data",
        "config",
    ):
        assert key in r, f"missing {key}"
    assert r["n_gpus"] == 1
    assert r["steps"] == 4 and r["warmup"] == 1
    assert r["higher_is_better"] is False
    assert r["scaling"] == "weak"
    assert r["unit"] == "ms"
    assert r["value"] > 0 and r["ms_per_step"] > 0
    assert "global_batch" in r["config"] and "parallelism" in r["config"]
