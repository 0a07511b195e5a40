"""Write-ahead log for MemoryMVCC.

One JSONL record per MVCC event:
``{"r": revision, "k": key, "v": value|null, "ver": version, "cr": create_rev}``.
Replayed in order at boot, this reconstructs the full store including history
(so rollback across daemon restarts works without an external etcd — the
reference instead requires etcd for exactly this).

``rewrite()`` compacts the file to the store's retained events (after
``MemoryMVCC.compact``) via write-temp + atomic rename; the explicit
``ver``/``cr`` fields keep per-key version numbering exact across a
compaction boundary.
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
        self._store: Optional[MemoryMVCC] = None
        self._records_since_check = 0
        # auto-rewrite threshold: ephemeral puts collapse in-memory history,
        # so a rewrite shrinks the log back to retained events
        self.max_bytes = 64 * 1024 * 1024

    def attach(self, store: MemoryMVCC) -> None:
        """Replay existing log into ``store``, then record its future events."""
        self._store = store
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
                    store.replay(
                        rec["k"],
                        int(rec["r"]),
                        rec.get("v"),
                        rec.get("ver"),
                        rec.get("cr"),
                    )
        self._f = open(self.path, "a", encoding="utf-8")
        store.on_event = self._record

    @staticmethod
    def _fmt(key: str, rev: int, value: Optional[str], version: int, create_rev: int) -> str:
        return json.dumps(
            {"r": rev, "k": key, "v": value, "ver": version, "cr": create_rev},
            separators=(",", ":"),
        )

    def _record(
        self, key: str, rev: int, value: Optional[str], version: int = 0, create_rev: int = 0
    ) -> None:
        assert self._f is not None
        self._f.write(self._fmt(key, rev, value, version, create_rev) + "\n")
        self._f.flush()
        if self.fsync:
            os.fsync(self._f.fileno())
        self._records_since_check += 1
        if self._records_since_check >= 8192:
            self._records_since_check = 0
            try:
                if os.path.getsize(self.path) > self.max_bytes:
                    self.rewrite()
            except OSError:
                pass

    def rewrite(self) -> int:
        """Rewrite the log with only the store's retained events. Returns the
        new file size in bytes."""
        assert self._store is not None
        tmp = self.path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            for key, rev, value, version, create_rev in self._store.dump_events():
                f.write(self._fmt(key, rev, value, version, create_rev) + "\n")
            f.flush()
            os.fsync(f.fileno())
        if self._f:
            self._f.close()
        os.replace(tmp, self.path)
        self._f = open(self.path, "a", encoding="utf-8")
        return os.path.getsize(self.path)

    def close(self) -> None:
        if self._f:
            self._f.close()
            self._f = None
