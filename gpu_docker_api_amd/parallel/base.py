"""Common scheduler scaffolding.

The reference declares (and never uses) a Scheduler interface
(/root/reference/internal/schedulers/scheduler.go:3-9). Here the base class
actually carries the shared behavior: a real lock around every mutation
(the reference's version maps are racy — SURVEY.md §7.4) and write-behind
persistence of the serialized state through the WorkQueue, with a
synchronous flush hook for shutdown (reference: main.go:144-151).
"""
from __future__ import annotations

import threading
from typing import Optional

from ..state.keys import Resource
from ..state.store import StateStore
from ..state.workqueue import WorkQueue


class BaseScheduler:
    resource: Resource
    state_key: str

    def __init__(self, store: StateStore, queue: Optional[WorkQueue]) -> None:
        self._store = store
        self._queue = queue
        self._lock = threading.RLock()

    def serialize(self) -> str:
        raise NotImplementedError

    def _persist_async(self) -> None:
        """Async write-behind (reference persists each mutation the same way,
        gpuscheduler.go:159-165 — except ports, which it forgets: a bug)."""
        if self._queue is not None:
            self._queue.put(self.resource, self.state_key, self.serialize())

    async def persist(self) -> None:
        """Synchronous persist (shutdown path)."""
        await self._store.put_ephemeral(self.resource, self.state_key, self.serialize())
