"""In-process MVCC key-value store with etcd v3 revision semantics.

This is the contract everything else leans on: the
reference's history/rollback features walk etcd per-key revisions
(/root/reference/internal/etcd/revision.go:18-66), so the semantics here must
match etcd exactly:

* A single store-wide **revision** counter starts at 1; every mutating
  transaction writes at ``current_revision + 1`` and advances the counter
  (so the first put lands at revision 2, as in etcd).
* Each key-value carries ``create_revision`` (revision at which the key was
  created in its current lifetime), ``mod_revision`` (revision of the last
  write) and ``version`` (number of writes since creation; resets when a key
  is deleted and re-created).
* ``get(key, rev=r)`` returns the key-value **as of revision r** — the state
  after all transactions with revision <= r — exactly what etcd's
  ``clientv3.WithRev`` does and what the reference's revision walker relies on.
* ``compact(rev)`` discards history below ``rev``; reads below the compaction
  point raise :class:`RevisionCompacted` (etcd: ErrCompacted).

Thread-safe (one ``RLock``); an optional write-ahead log (``wal.py``) makes it
durable so a single-node deployment needs no external etcd.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional

from ..xerrors import NotExistInStore, RevisionCompacted


@dataclass(frozen=True)
class KeyValue:
    """One key-value as etcd would return it."""

    key: str
    value: str
    create_revision: int
    mod_revision: int
    version: int

    def to_dict(self) -> dict:
        return {
            "key": self.key,
            "value": self.value,
            "create_revision": self.create_revision,
            "mod_revision": self.mod_revision,
            "version": self.version,
        }


@dataclass(frozen=True)
class _Event:
    """One write to a key: a put (value is str) or a tombstone (value None)."""

    mod_revision: int
    value: Optional[str]
    version: int          # etcd per-key Version at this event (0 for tombstone)
    create_revision: int  # lifetime start (0 for tombstone)


class MemoryMVCC:
    """The store. Mutations may be observed via ``on_event`` (used by the WAL)."""

    def __init__(self, prune_dead_lifetimes: bool = False) -> None:
        self._lock = threading.RLock()
        self._rev = 1            # etcd: store starts at revision 1
        self._compacted = 0      # highest compacted revision (exclusive floor)
        # when True, re-creating a deleted key drops the dead lifetime's
        # events (only at-revision reads of dead lifetimes could see them;
        # nothing in the daemon does). Bounds churn like delete+recreate
        # loops; etcd's answer to the same growth is compaction.
        self.prune_dead_lifetimes = prune_dead_lifetimes
        self._hist: Dict[str, List[_Event]] = {}
        # observers(key, rev, value, version, create_revision) — value None
        # for tombstones. Version/create_revision ride along so a WAL
        # rewritten after compaction replays with exact per-key numbering.
        # Multiple observers: the WAL plus any event-stream subscribers.
        self.observers: List[Callable[[str, int, Optional[str], int, int], None]] = []

    @property
    def on_event(self):  # backward-compat single-observer view
        return self.observers[0] if self.observers else None

    @on_event.setter
    def on_event(self, cb) -> None:
        if cb is not None:
            self.observers.append(cb)

    def _notify(self, key, rev, value, version, create_rev) -> None:
        for cb in self.observers:
            cb(key, rev, value, version, create_rev)

    # ------------------------------------------------------------------ info
    @property
    def revision(self) -> int:
        with self._lock:
            return self._rev

    @property
    def compacted_revision(self) -> int:
        with self._lock:
            return self._compacted

    # ----------------------------------------------------------------- write
    def put(self, key: str, value: str, retain_history: bool = True) -> KeyValue:
        """Write a key. ``retain_history=False`` replaces the key's previous
        put event instead of appending — for high-churn singleton state
        (scheduler bitmaps, version maps) whose history nobody walks; without
        it those keys grow the store without bound (etcd has the same issue
        and answers with compaction). Revision/version counters still advance
        so readers can't tell the difference at the head.
        """
        with self._lock:
            rev = self._rev + 1
            events = self._hist.setdefault(key, [])
            last = events[-1] if events else None
            if last is None or last.value is None:
                create_rev, version = rev, 1
                if last is not None and self.prune_dead_lifetimes:
                    events[:] = [last]  # keep only the tombstone
            else:
                create_rev, version = last.create_revision, last.version + 1
                if not retain_history:
                    events.pop()  # collapse: at most one live event retained
            events.append(_Event(rev, value, version, create_rev))
            self._rev = rev
            self._notify(key, rev, value, version, create_rev)
            return KeyValue(key, value, create_rev, rev, version)

    def delete(self, key: str) -> int:
        """Delete a key. Returns the number of keys deleted (0 or 1).

        A successful delete advances the revision and leaves a tombstone, so
        ``get(key, rev=old)`` still sees pre-delete values (etcd behavior).
        """
        with self._lock:
            events = self._hist.get(key)
            if not events or events[-1].value is None:
                return 0
            rev = self._rev + 1
            events.append(_Event(rev, None, 0, 0))
            self._rev = rev
            self._notify(key, rev, None, 0, 0)
            return 1

    def delete_prefix(self, prefix: str) -> int:
        """Delete all live keys under a prefix in ONE transaction (etcd
        DeleteRange semantics: a single revision bump for the whole range)."""
        with self._lock:
            live = [
                k
                for k, ev in self._hist.items()
                if k.startswith(prefix) and ev and ev[-1].value is not None
            ]
            if not live:
                return 0
            rev = self._rev + 1
            for k in sorted(live):
                self._hist[k].append(_Event(rev, None, 0, 0))
                self._notify(k, rev, None, 0, 0)
            self._rev = rev
            return len(live)

    # ------------------------------------------------------------------ read
    def get(self, key: str, rev: int = 0) -> KeyValue:
        """Latest value (rev=0) or value as of revision ``rev``.

        Raises NotExistInStore if absent (at that revision), RevisionCompacted
        if ``rev`` is at or below the compaction point.
        """
        with self._lock:
            kv = self._get_locked(key, rev)
            if kv is None:
                raise NotExistInStore(key)
            return kv

    def get_or_none(self, key: str, rev: int = 0) -> Optional[KeyValue]:
        with self._lock:
            return self._get_locked(key, rev)

    def _get_locked(self, key: str, rev: int) -> Optional[KeyValue]:
        if rev:
            if rev <= self._compacted:
                raise RevisionCompacted(f"revision {rev} compacted at {self._compacted}")
            if rev > self._rev:
                # etcd returns ErrFutureRev; surface as missing-with-context
                raise NotExistInStore(f"{key}@{rev} (future revision, store at {self._rev})")
        events = self._hist.get(key)
        if not events:
            return None
        if rev == 0:
            ev = events[-1]
        else:
            ev = None
            for e in reversed(events):
                if e.mod_revision <= rev:
                    ev = e
                    break
            if ev is None:
                return None
        if ev.value is None:
            return None
        return KeyValue(key, ev.value, ev.create_revision, ev.mod_revision, ev.version)

    def range_prefix(self, prefix: str, rev: int = 0) -> List[KeyValue]:
        """All live key-values under a prefix (as of ``rev`` if nonzero),
        sorted by key (etcd Range default order)."""
        with self._lock:
            out: List[KeyValue] = []
            for key in sorted(self._hist):
                if not key.startswith(prefix):
                    continue
                kv = self._get_locked(key, rev)
                if kv is not None:
                    out.append(kv)
            return out

    # -------------------------------------------------------------- history
    def history(self, key: str) -> List[KeyValue]:
        """All surviving put-revisions of a key, newest first.

        Matches what the reference's walker extracts: it walks rev from
        mod_revision down to create_revision issuing one Get(WithRev) per
        revision and dedupes by per-key Version (etcd/revision.go:18-44).
        We return the same set directly — every put event of the key's
        current lifetime that is above the compaction point. O(#events),
        not O(#store revisions): this is the "hot loop" the reference made
        O(total revisions) round-trips for.
        """
        with self._lock:
            events = self._hist.get(key)
            if not events or events[-1].value is None:
                raise NotExistInStore(key)
            create_rev = events[-1].create_revision
            lifetime = [e for e in events if e.value is not None and e.create_revision == create_rev]
            # events above the compaction floor, PLUS the floor itself: the
            # state at the compaction revision stays readable in etcd
            # (Get(WithRev=compacted+1) returns it), so the walker would
            # surface it too.
            floor = None
            out_events = []
            for e in lifetime:
                if e.mod_revision > self._compacted:
                    out_events.append(e)
                else:
                    floor = e
            if floor is not None:
                out_events.insert(0, floor)
            return [
                KeyValue(key, e.value, e.create_revision, e.mod_revision, e.version)
                for e in reversed(out_events)
            ]

    def get_version(self, key: str, version: int) -> KeyValue:
        """The key-value whose per-key Version equals ``version`` (current
        lifetime) — the reference's GetRevision(version) walk
        (etcd/revision.go:46-66) without the O(revisions) round trips."""
        for kv in self.history(key):
            if kv.version == version:
                return kv
        raise NotExistInStore(f"{key} version={version}")

    # ------------------------------------------------------------ compaction
    def compact(self, rev: int) -> None:
        """Discard event history with mod_revision < rev (etcd keeps the
        latest event at/below the compaction point per key)."""
        with self._lock:
            if rev > self._rev:
                raise NotExistInStore(f"cannot compact future revision {rev}")
            if rev <= self._compacted:
                raise RevisionCompacted(f"already compacted at {self._compacted}")
            for key, events in list(self._hist.items()):
                keep: List[_Event] = []
                # keep the newest event <= rev (it defines state at rev) ...
                floor = None
                for e in events:
                    if e.mod_revision <= rev:
                        floor = e
                    else:
                        keep.append(e)
                if floor is not None and (floor.value is not None or keep):
                    keep.insert(0, floor)
                if keep:
                    self._hist[key] = keep
                else:
                    del self._hist[key]
            self._compacted = rev - 1 if rev > 0 else 0

    # ---------------------------------------------------------------- replay
    def replay(
        self,
        key: str,
        rev: int,
        value: Optional[str],
        version: Optional[int] = None,
        create_rev: Optional[int] = None,
    ) -> None:
        """Re-apply a WAL record. Only valid in ascending rev order.
        Explicit version/create_rev (post-compaction WALs) take precedence
        over the computed lifetime numbering."""
        with self._lock:
            events = self._hist.setdefault(key, [])
            last = events[-1] if events else None
            if value is None:
                events.append(_Event(rev, None, 0, 0))
            else:
                if version is None or create_rev is None:
                    if last is None or last.value is None:
                        create_rev, version = rev, 1
                    else:
                        create_rev, version = last.create_revision, last.version + 1
                events.append(_Event(rev, value, version, create_rev))
            if rev > self._rev:
                self._rev = rev

    def dump_events(self):
        """All retained events in revision order (WAL rewrite after
        compaction): (key, rev, value|None, version, create_revision)."""
        with self._lock:
            out = []
            for key, events in self._hist.items():
                for e in events:
                    out.append((key, e.mod_revision, e.value, e.version, e.create_revision))
            out.sort(key=lambda t: t[1])
            return out
