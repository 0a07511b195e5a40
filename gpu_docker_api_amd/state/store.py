"""StateStore — the async facade services use.

Two backends:

* :class:`MemoryStore` — in-process MVCC (+ optional WAL), the default; a
  single-node daemon needs no external services.
* :class:`EtcdGatewayStore` (``etcd_gateway.py``) — real etcd over its v3
  JSON gateway, same key scheme and value shapes as the reference, so an
  etcd populated by either implementation is readable by the other.

API mirrors what the reference layers on clientv3
(/root/reference/internal/etcd/common.go:45-68, revision.go:18-66):
put / get / delete per (resource, key), plus history and get_version which
replace the reference's O(total-revisions) walker with an O(#writes) path.
"""
from __future__ import annotations

from typing import List, Optional

from .keys import Resource, resource_key, resource_prefix
from .mvcc import KeyValue, MemoryMVCC
from .wal import Wal


class StateStore:
    """Interface. All methods are coroutines so network backends fit."""

    async def put(self, resource: Resource, key: str, value: str) -> None:
        raise NotImplementedError

    async def put_ephemeral(self, resource: Resource, key: str, value: str) -> None:
        """Write WITHOUT retaining history — for the high-churn singleton
        state keys (scheduler bitmaps, version maps). Default: plain put
        (real etcd keeps revisions until compacted, as with the reference)."""
        await self.put(resource, key, value)

    async def get(self, resource: Resource, key: str) -> KeyValue:
        """Raises NotExistInStore when absent."""
        raise NotImplementedError

    async def get_or_none(self, resource: Resource, key: str) -> Optional[KeyValue]:
        from ..xerrors import NotExistInStore

        try:
            return await self.get(resource, key)
        except NotExistInStore:
            return None

    async def delete(self, resource: Resource, key: str) -> int:
        raise NotImplementedError

    async def range(self, resource: Resource) -> List[KeyValue]:
        raise NotImplementedError

    async def history(self, resource: Resource, key: str) -> List[KeyValue]:
        """All surviving versions of a key, newest first."""
        raise NotImplementedError

    async def get_version(self, resource: Resource, key: str, version: int) -> KeyValue:
        raise NotImplementedError

    async def close(self) -> None:
        pass


class MemoryStore(StateStore):
    def __init__(self, wal_path: Optional[str] = None, fsync: bool = False) -> None:
        self.mvcc = MemoryMVCC(prune_dead_lifetimes=True)
        self._wal: Optional[Wal] = None
        if wal_path:
            self._wal = Wal(wal_path, fsync=fsync)
            self._wal.attach(self.mvcc)
        # live event-stream subscribers: (asyncio queue, its loop)
        self._subscribers: list = []
        self.mvcc.observers.append(self._fanout)

    def _fanout(self, key: str, rev: int, value, version: int, create_rev: int) -> None:
        if not self._subscribers:
            return
        parts = key.rsplit("/", 2)
        event = {
            "type": "delete" if value is None else "put",
            "key": key,
            "resource": parts[-2] if len(parts) >= 2 else "",
            "name": parts[-1],
            "revision": rev,
            "version": version,
        }
        for q, loop in list(self._subscribers):
            try:  # observers may fire from any thread; hop to the queue's loop
                loop.call_soon_threadsafe(q.put_nowait, event)
            except RuntimeError:
                pass

    def subscribe(self):
        """Returns (queue, unsubscribe) for the live state-change stream."""
        import asyncio

        q: asyncio.Queue = asyncio.Queue(maxsize=1024)
        loop = asyncio.get_running_loop()
        entry = (q, loop)
        self._subscribers.append(entry)

        def unsubscribe() -> None:
            try:
                self._subscribers.remove(entry)
            except ValueError:
                pass

        return q, unsubscribe

    async def put(self, resource: Resource, key: str, value: str) -> None:
        self.mvcc.put(resource_key(resource, key), value)

    async def put_ephemeral(self, resource: Resource, key: str, value: str) -> None:
        self.mvcc.put(resource_key(resource, key), value, retain_history=False)

    async def get(self, resource: Resource, key: str) -> KeyValue:
        return self.mvcc.get(resource_key(resource, key))

    async def delete(self, resource: Resource, key: str) -> int:
        return self.mvcc.delete(resource_key(resource, key))

    async def range(self, resource: Resource) -> List[KeyValue]:
        return self.mvcc.range_prefix(resource_prefix(resource) + "/")

    async def history(self, resource: Resource, key: str) -> List[KeyValue]:
        return self.mvcc.history(resource_key(resource, key))

    async def get_version(self, resource: Resource, key: str, version: int) -> KeyValue:
        return self.mvcc.get_version(resource_key(resource, key), version)

    async def compact(self, revision: int = 0) -> dict:
        """Discard history below ``revision`` (default: everything but the
        live state) and shrink the WAL. History queries below the point
        raise RevisionCompacted afterwards — an explicit operator action."""
        rev = revision or self.mvcc.revision
        self.mvcc.compact(rev)
        wal_bytes = self._wal.rewrite() if self._wal else 0
        return {"compacted_revision": rev, "wal_bytes": wal_bytes}

    async def close(self) -> None:
        if self._wal:
            self._wal.close()
