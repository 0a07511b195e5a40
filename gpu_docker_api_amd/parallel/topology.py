"""xGMI topology model + placement scoring.

The reference's GPU selection is first-fit over random map iteration with no
topology awareness (/root/reference/internal/schedulers/gpuscheduler.go:93-102).
On an MI355X node that is the wrong model: the 8 GPUs are connected
point-to-point by xGMI (7 links per GPU, ≈153 GB/s each) and ring collectives
are per-link bound, so a multi-GPU allocation should maximize intra-set link
bandwidth. This module holds the adjacency matrix — a static estimate from
amdsmi link types, overlaid by *measured* numbers from the native HIP
bandwidth probe (csrc/hipcore.hip p2p_matrix via ops.hipcore) when available — and
scores candidate GPU subsets for the scheduler's bin-pack.
"""
from __future__ import annotations

import itertools
import json
import logging
import os
from typing import Dict, List, Optional, Sequence

log = logging.getLogger(__name__)


class Topology:
    def __init__(self, matrix: List[List[float]], uuids: List[str]) -> None:
        assert len(matrix) == len(uuids)
        self.matrix = matrix
        self.uuids = uuids
        self.index_of: Dict[str, int] = {u: i for i, u in enumerate(uuids)}
        self.measured = False  # True once the HIP probe overlaid real numbers

    @property
    def n(self) -> int:
        return len(self.uuids)

    def bandwidth(self, a: str, b: str) -> float:
        return self.matrix[self.index_of[a]][self.index_of[b]]

    # ---------------------------------------------------------------- probe
    def overlay_measured(self, probe: dict) -> None:
        """Merge the probe's output: {"gpus": [uuid...], "p2p_gbps": [[...]]}
        (self-bandwidth on the diagonal is ignored for placement)."""
        uuids = probe.get("gpus") or []
        mat = probe.get("p2p_gbps") or []
        for i, ui in enumerate(uuids):
            for j, uj in enumerate(uuids):
                if i == j or ui not in self.index_of or uj not in self.index_of:
                    continue
                v = float(mat[i][j])
                if v > 0:
                    self.matrix[self.index_of[ui]][self.index_of[uj]] = v
        self.measured = True

    @classmethod
    def load_probe_file(cls, path: str) -> Optional[dict]:
        if not os.path.exists(path):
            return None
        try:
            with open(path) as f:
                return json.load(f)
        except Exception as exc:
            log.warning("bad probe file %s: %s", path, exc)
            return None

    # ------------------------------------------------------------- placement
    def subset_score(self, indices: Sequence[int]) -> float:
        """Score of a candidate set: the minimum pairwise bandwidth (ring
        collectives are bound by the weakest link), tie-broken by the sum."""
        if len(indices) < 2:
            return float("inf")
        pairs = list(itertools.combinations(indices, 2))
        bws = [self.matrix[a][b] for a, b in pairs]
        return min(bws) * 1e6 + sum(bws)

    def best_subset(self, free: Sequence[int], n: int, exhaustive_limit: int = 70) -> List[int]:
        """Pick n GPUs from ``free`` maximizing subset_score.

        Exhaustive when C(len(free), n) is small (it always is on one 8-GPU
        node: C(8,4)=70); greedy seeded by the best-connected pair otherwise.
        """
        free = sorted(free)
        if n >= len(free):
            return list(free)
        if n <= 1:
            # prefer the lowest index for determinism
            return [free[0]]
        ncomb = 1
        k = min(n, len(free) - n)
        for i in range(k):
            ncomb = ncomb * (len(free) - i) // (i + 1)
        if ncomb <= exhaustive_limit:
            best = max(
                itertools.combinations(free, n), key=lambda c: (self.subset_score(c), [-i for i in c])
            )
            return sorted(best)
        # greedy: best pair, then add the GPU maximizing the running score
        best_pair = max(
            itertools.combinations(free, 2), key=lambda c: self.matrix[c[0]][c[1]]
        )
        chosen = list(best_pair)
        remaining = [i for i in free if i not in chosen]
        while len(chosen) < n:
            nxt = max(remaining, key=lambda r: self.subset_score(chosen + [r]))
            chosen.append(nxt)
            remaining.remove(nxt)
        return sorted(chosen)

    def to_dict(self) -> dict:
        return {
            "gpus": self.uuids,
            "linkGbps": [[round(v, 1) for v in row] for row in self.matrix],
            "measured": self.measured,
        }
