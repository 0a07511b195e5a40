"""Explicit undo-stack for multi-step operations.

The reference's rolling-replace flows have ~7 failure points spanning two
containers, three schedulers and the state store, unwound by hand-written
``if err != nil { restore... }`` chains that are provably incomplete
(e.g. version bumped before etcd flush,
services/replicaset.go:102-147). Here every step that acquires a resource
pushes its compensation; on failure compensations run LIFO, each isolated so
one failing undo doesn't strand the rest.
"""
from __future__ import annotations

import inspect
import logging
from typing import Awaitable, Callable, List, Union

log = logging.getLogger(__name__)

Undo = Callable[[], Union[None, Awaitable[None]]]


class Saga:
    def __init__(self, name: str = "") -> None:
        self.name = name
        self._undos: List[tuple[str, Undo]] = []
        self._committed = False

    def push(self, label: str, undo: Undo) -> None:
        self._undos.append((label, undo))

    def commit(self) -> None:
        """Operation succeeded: discard compensations."""
        self._committed = True
        self._undos.clear()

    async def rollback(self) -> None:
        if self._committed:
            return
        while self._undos:
            label, undo = self._undos.pop()
            try:
                result = undo()
                if inspect.isawaitable(result):
                    await result
            except Exception:  # noqa: BLE001 — keep unwinding
                log.exception("saga %s: compensation %r failed", self.name, label)

    async def __aenter__(self) -> "Saga":
        return self

    async def __aexit__(self, exc_type, exc, tb) -> bool:
        if exc_type is not None:
            await self.rollback()
        return False
