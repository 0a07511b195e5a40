"""Async write-behind persistence queue.

The reference buffers Put/Del messages on a channel and a SyncLoop goroutine
flushes them to etcd, re-enqueueing forever on failure
(/root/reference/internal/workQueue/workQueue.go:12-54). Differences here:

* retries are bounded with exponential backoff (unbounded retry of a
  poisoned record would wedge the reference's loop);
* ``drain()`` lets shutdown and tests wait for all pending writes — the
  reference's shutdown can lose queued writes;
* the queue size is configurable (reference hardcodes 110).
"""
from __future__ import annotations

import asyncio
import logging
from dataclasses import dataclass, field

from .keys import Resource
from .store import StateStore

log = logging.getLogger(__name__)


@dataclass
class PutKeyValue:
    resource: Resource
    key: str
    value: str
    attempts: int = field(default=0, compare=False)


@dataclass
class DelKey:
    resource: Resource
    key: str
    attempts: int = field(default=0, compare=False)


class WorkQueue:
    def __init__(self, store: StateStore, maxsize: int = 1024, max_attempts: int = 8) -> None:
        self.store = store
        self.max_attempts = max_attempts
        self._q: asyncio.Queue = asyncio.Queue(maxsize=maxsize)
        self._task: asyncio.Task | None = None
        self._closing = False
        # items sitting in a retry backoff: neither in the queue nor done,
        # so drain() must wait for them too (an etcd outage would otherwise
        # let shutdown proceed while writes are still pending)
        self._retrying = 0

    def start(self) -> None:
        if self._task is None:
            self._task = asyncio.get_running_loop().create_task(self._sync_loop())

    def enqueue(self, item: PutKeyValue | DelKey) -> None:
        if self._closing:
            return
        try:
            self._q.put_nowait(item)
        except asyncio.QueueFull:
            # back-pressure rather than dropping state writes
            log.warning("workqueue full; applying synchronously is not possible here, blocking")
            asyncio.get_running_loop().create_task(self._q.put(item))

    def put(self, resource: Resource, key: str, value: str) -> None:
        self.enqueue(PutKeyValue(resource, key, value))

    def delete(self, resource: Resource, key: str) -> None:
        self.enqueue(DelKey(resource, key))

    async def _apply(self, item: PutKeyValue | DelKey) -> None:
        if isinstance(item, PutKeyValue):
            # everything routed through the queue is singleton scheduler /
            # version-map state: ephemeral (no history retention) — a 250
            # ops/s daemon would otherwise grow the store without bound
            await self.store.put_ephemeral(item.resource, item.key, item.value)
        else:
            await self.store.delete(item.resource, item.key)

    async def _sync_loop(self) -> None:
        while True:
            item = await self._q.get()
            try:
                if item is None:
                    return
                try:
                    await self._apply(item)
                except Exception as exc:  # noqa: BLE001 — persistence must not die
                    item.attempts += 1
                    if item.attempts >= self.max_attempts:
                        log.error("workqueue: dropping %r after %d attempts: %s", item, item.attempts, exc)
                    else:
                        delay = min(0.05 * (2 ** item.attempts), 5.0)
                        log.warning("workqueue: retrying %r in %.2fs: %s", item, delay, exc)
                        loop = asyncio.get_running_loop()
                        self._retrying += 1
                        loop.call_later(delay, self._requeue, item)
            finally:
                self._q.task_done()

    def _requeue(self, item) -> None:
        self._retrying -= 1
        self.enqueue(item)

    async def drain(self) -> None:
        """Wait until everything enqueued OR in a retry backoff is flushed
        (or dropped at max_attempts)."""
        while True:
            await self._q.join()
            if self._retrying == 0 and self._q.empty():
                return
            await asyncio.sleep(0.05)

    async def close(self) -> None:
        await self.drain()
        self._closing = True
        await self._q.join()
        if self._task is not None:
            await self._q.put(None)
            await self._task
            self._task = None
