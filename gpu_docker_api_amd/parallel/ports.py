"""Host-port scheduler.

Parity with the reference (portscheduler.go:76-161): Apply(n) picks n free
ports from the configured range (default 40000-65535, main.go:36), Restore
frees them, status reports start/end/used set. State persists under
``ports/usedPortSetKey`` with the reference's Go JSON shape
``{"StartPort":..,"EndPort":..,"AvailableCount":..,"UsedPortSet":{"p":{}}}``.

Fixes two reference bugs: (a) its per-mutation persistence wrote the *GPU*
map to the gpus key (portscheduler.go:163-169) so port state only reached
etcd at shutdown — here every mutation persists the port set; (b) its random
probe loop is unbounded — here probing is bounded and falls back to a scan.
"""
from __future__ import annotations

import json
import random
from typing import Dict, List, Optional, Set

from ..state.keys import USED_PORT_SET_KEY, Resource
from ..state.store import StateStore
from ..state.workqueue import WorkQueue
from ..xerrors import PortNotEnough
from .base import BaseScheduler


class PortScheduler(BaseScheduler):
    resource = Resource.PORTS
    state_key = USED_PORT_SET_KEY

    def __init__(
        self,
        store: StateStore,
        queue: Optional[WorkQueue],
        start_port: int = 40000,
        end_port: int = 65535,
    ) -> None:
        super().__init__(store, queue)
        self.start_port = start_port
        self.end_port = end_port
        self.available_count = end_port - start_port + 1
        self.used: Set[int] = set()
        self._rng = random.Random()

    @classmethod
    async def create(
        cls,
        store: StateStore,
        queue: Optional[WorkQueue],
        start_port: int = 40000,
        end_port: int = 65535,
    ) -> "PortScheduler":
        self = cls(store, queue, start_port, end_port)
        kv = await store.get_or_none(self.resource, self.state_key)
        if kv is not None:
            data = json.loads(kv.value)
            self.used = {
                int(p)
                for p in (data.get("UsedPortSet") or {})
                if self.start_port <= int(p) <= self.end_port
            }
        else:
            await self.persist()
        return self

    def serialize(self) -> str:
        with self._lock:
            return json.dumps(
                {
                    "StartPort": self.start_port,
                    "EndPort": self.end_port,
                    "AvailableCount": self.available_count,
                    "UsedPortSet": {str(p): {} for p in sorted(self.used)},
                },
                separators=(",", ":"),
            )

    def apply(self, num: int) -> List[int]:
        if num <= 0 or num > self.available_count:
            raise PortNotEnough(f"requested {num}, range has {self.available_count}")
        with self._lock:
            free_count = self.available_count - len(self.used)
            if free_count < num:
                raise PortNotEnough(f"requested {num}, only {free_count} free")
            chosen: List[int] = []
            span = self.end_port - self.start_port + 1
            attempts = 0
            while len(chosen) < num and attempts < num * 16:
                p = self._rng.randrange(span) + self.start_port
                attempts += 1
                if p not in self.used:
                    self.used.add(p)
                    chosen.append(p)
            if len(chosen) < num:  # dense range: deterministic scan
                for p in range(self.start_port, self.end_port + 1):
                    if p not in self.used:
                        self.used.add(p)
                        chosen.append(p)
                        if len(chosen) == num:
                            break
            self._persist_async()
            return chosen

    def apply_specific(self, ports: List[int] | List[str]) -> None:
        """Re-acquire exact host ports (startup of a stopped container)."""
        ints = [int(p) for p in ports]
        if not ints:
            return
        with self._lock:
            busy = [p for p in ints if p in self.used]
            if busy:
                raise PortNotEnough(f"ports already allocated: {busy}")
            for p in ints:
                self.used.add(p)
            self._persist_async()

    def restore(self, ports: List[int] | List[str]) -> None:
        if not ports:
            return
        with self._lock:
            for p in ports:
                self.used.discard(int(p))
            self._persist_async()

    def get_port_status(self) -> Dict:
        with self._lock:
            return {
                "StartPort": self.start_port,
                "EndPort": self.end_port,
                "AvailableCount": self.available_count - len(self.used),
                "UsedPortSet": {str(p): {} for p in sorted(self.used)},
            }
