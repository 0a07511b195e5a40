"""Phase-timestamped latency accounting.

The headline metric of this control plane IS operation latency (BASELINE.json:
p50 create->running and patch turnaround), so every service operation records
its phases (schedule, create, start, copy, persist). The reference has no
tracing at all. Exposed via GET /metrics (Prometheus text)
and attached to responses when requested.
"""
from __future__ import annotations

import threading
import time
from collections import defaultdict
from typing import Dict, List


class PhaseTimer:
    def __init__(self, op: str) -> None:
        self.op = op
        self.t0 = time.perf_counter()
        self._last = self.t0
        self.phases: List[tuple[str, float]] = []

    def mark(self, phase: str) -> None:
        now = time.perf_counter()
        self.phases.append((phase, (now - self._last) * 1000.0))
        self._last = now

    @property
    def total_ms(self) -> float:
        return (time.perf_counter() - self.t0) * 1000.0

    def to_dict(self) -> Dict[str, float]:
        d = {f"{name}_ms": round(ms, 3) for name, ms in self.phases}
        d["total_ms"] = round(self.total_ms, 3)
        return d

    def finish(self) -> Dict[str, float]:
        METRICS.observe(self.op, self.total_ms)
        for name, ms in self.phases:
            METRICS.observe(f"{self.op}.{name}", ms)
        return self.to_dict()


class Metrics:
    """Tiny in-process histogram/counter registry (p50/p95/p99 per op)."""

    def __init__(self, keep: int = 4096) -> None:
        self._lock = threading.Lock()
        self._keep = keep
        self._samples: Dict[str, List[float]] = defaultdict(list)
        self._counts: Dict[str, int] = defaultdict(int)

    def observe(self, name: str, ms: float) -> None:
        with self._lock:
            self._counts[name] += 1
            s = self._samples[name]
            s.append(ms)
            if len(s) > self._keep:
                del s[: len(s) - self._keep]

    def count(self, name: str) -> None:
        with self._lock:
            self._counts[name] += 1

    @staticmethod
    def _pct(sorted_samples: List[float], q: float) -> float:
        if not sorted_samples:
            return 0.0
        idx = min(int(q * len(sorted_samples)), len(sorted_samples) - 1)
        return sorted_samples[idx]

    def summary(self) -> Dict[str, dict]:
        with self._lock:
            out = {}
            for name, s in self._samples.items():
                ss = sorted(s)
                out[name] = {
                    "count": self._counts[name],
                    "p50_ms": round(self._pct(ss, 0.50), 3),
                    "p95_ms": round(self._pct(ss, 0.95), 3),
                    "p99_ms": round(self._pct(ss, 0.99), 3),
                    "mean_ms": round(sum(ss) / len(ss), 3) if ss else 0.0,
                }
            for name, c in self._counts.items():
                if name not in out:
                    out[name] = {"count": c}
            return out

    def prometheus_text(self) -> str:
        lines = []
        for name, st in sorted(self.summary().items()):
            metric = "gda_" + name.replace(".", "_").replace("-", "_")
            lines.append(f"{metric}_count {st.get('count', 0)}")
            for k in ("p50_ms", "p95_ms", "p99_ms", "mean_ms"):
                if k in st:
                    lines.append(f"{metric}_{k} {st[k]}")
        return "\n".join(lines) + "\n"

    def reset(self) -> None:
        with self._lock:
            self._samples.clear()
            self._counts.clear()


METRICS = Metrics()
