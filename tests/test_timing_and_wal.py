"""Direct coverage for the metrics registry, phase timers and WAL fsync."""
import os

from gpu_docker_api_amd.state.mvcc import MemoryMVCC
from gpu_docker_api_amd.state.wal import Wal
from gpu_docker_api_amd.utils.timing import Metrics, PhaseTimer


def test_phase_timer_records_phases_and_metrics():
    m = Metrics()
    import gpu_docker_api_amd.utils.timing as timing

    orig = timing.METRICS
    timing.METRICS = m
    try:
        t = PhaseTimer("op.test")
        t.mark("alpha")
        t.mark("beta")
        d = t.finish()
        assert set(d) == {"alpha_ms", "beta_ms", "total_ms"}
        assert d["total_ms"] >= d["alpha_ms"]
        summary = m.summary()
        assert summary["op.test"]["count"] == 1
        assert "op.test.alpha" in summary and "op.test.beta" in summary
    finally:
        timing.METRICS = orig


def test_metrics_percentiles_and_cap():
    m = Metrics(keep=100)
    for i in range(1000):
        m.observe("lat", float(i))
    s = m.summary()["lat"]
    assert s["count"] == 1000
    # only the newest 100 samples retained: percentiles from [900, 999]
    assert 900 <= s["p50_ms"] <= 999
    assert s["p99_ms"] >= s["p95_ms"] >= s["p50_ms"]
    text = m.prometheus_text()
    assert "gda_lat_count 1000" in text


def test_wal_fsync_mode(tmp_path):
    p = str(tmp_path / "wal")
    s = MemoryMVCC()
    w = Wal(p, fsync=True)
    w.attach(s)
    s.put("/k", "v1")
    s.put("/k", "v2")
    w.close()
    s2 = MemoryMVCC()
    Wal(p, fsync=True).attach(s2)
    assert s2.get("/k").value == "v2"
    assert s2.get("/k").version == 2
    assert os.path.getsize(p) > 0
