"""Write-ahead log for MemoryMVCC.

One JSONL record per MVCC event: {"r": revision, "k": key, "v": value|null}.
Replayed in order at boot, this reconstructs the full store including history
(so rollback across daemon restarts works without an external etcd — the
reference instead requires etcd for exactly this).
"""
from __future__ import annotations

import json
import os
from typing import Optional

from .mvcc import MemoryMVCC


class Wal:
    def __init__(self, path: str, fsync: bool = False) -> None:
        self.path = path
        self.fsync = fsync
        os.makedirs(os.path.dirname(os.path.abspath(path)) or ".", exist_ok=True)
        self._f = None

    def attach(self, store: MemoryMVCC) -> None:
        """Replay existing log into ``store``, then record its future events."""
        if os.path.exists(self.path):
            with open(self.path, "r", encoding="utf-8") as f:
                for line in f:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        rec = json.loads(line)
                    except json.JSONDecodeError:
                        # torn tail write from a crash: stop replay there
                        break
                    store.replay(rec["k"], int(rec["r"]), rec.get("v"))
        self._f = open(self.path, "a", encoding="utf-8")
        store.on_event = self._record

    def _record(self, key: str, rev: int, value: Optional[str]) -> None:
        assert self._f is not None
        self._f.write(json.dumps({"r": rev, "k": key, "v": value}, separators=(",", ":")) + "\n")
        self._f.flush()
        if self.fsync:
            os.fsync(self._f.fileno())

    def close(self) -> None:
        if self._f:
            self._f.close()
            self._f = None
