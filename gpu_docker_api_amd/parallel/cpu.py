"""CPU scheduler: cpuset bitmap allocator.

Parity with the reference (cpuscheduler.go:77-159): Apply(n) returns a
comma-joined cpuset string for HostConfig.CpusetCpus, Restore frees the ids,
state persists under ``cpus/cpuStatusMapKey`` as
``{"availableCpuNums": N, "cpuStatusMap": {"0": 0|1, ...}}``.

Fixes the reference's allocation bug (cpuscheduler.go:94: iterating slice
*indices* instead of sorted cpu ids — only accidentally correct for
contiguous ids): we sort numerically and allocate the lowest free ids.

MI355X extension: ``apply(n, preferred_nodes=[...])`` prefers CPUs on the
NUMA node(s) of the container's GPUs (sysfs cpu->node map), so
host<->device staging stays socket-local; falls back to any free CPUs.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional

from ..state.keys import CPU_STATUS_MAP_KEY, Resource
from ..state.store import StateStore
from ..state.workqueue import WorkQueue
from ..xerrors import CpuNotEnough
from .base import BaseScheduler


class CpuScheduler(BaseScheduler):
    resource = Resource.CPUS
    state_key = CPU_STATUS_MAP_KEY

    def __init__(
        self,
        store: StateStore,
        queue: Optional[WorkQueue],
        count: int = 0,
        cpu_nodes: Optional[Dict[int, int]] = None,
    ) -> None:
        super().__init__(store, queue)
        self.available_cpu_nums = count or (os.cpu_count() or 1)
        self.cpu_status_map: Dict[str, int] = {}
        if cpu_nodes is None:
            from .numa import cpu_node_map

            cpu_nodes = cpu_node_map()
        self.cpu_nodes: Dict[int, int] = cpu_nodes or {}

    @classmethod
    async def create(
        cls,
        store: StateStore,
        queue: Optional[WorkQueue],
        count: int = 0,
        cpu_nodes: Optional[Dict[int, int]] = None,
    ) -> "CpuScheduler":
        self = cls(store, queue, count, cpu_nodes)
        kv = await store.get_or_none(self.resource, self.state_key)
        if kv is not None:
            data = json.loads(kv.value)
            persisted = data.get("cpuStatusMap") or {}
            self.cpu_status_map = {
                str(i): int(persisted.get(str(i), 0)) for i in range(self.available_cpu_nums)
            }
        else:
            self.cpu_status_map = {str(i): 0 for i in range(self.available_cpu_nums)}
            await self.persist()
        return self

    def serialize(self) -> str:
        with self._lock:
            return json.dumps(
                {
                    "availableCpuNums": self.available_cpu_nums,
                    "cpuStatusMap": self.cpu_status_map,
                },
                separators=(",", ":"),
            )

    def apply(self, num: int, preferred_nodes: Optional[List[int]] = None) -> str:
        """Allocate ``num`` cpus; returns 'i,j,k' for CpusetCpus.
        ``preferred_nodes``: NUMA nodes to draw from first (the allocated
        GPUs' nodes) — spills to other nodes only when they run dry."""
        if num <= 0 or num > self.available_cpu_nums:
            raise CpuNotEnough(f"requested {num}, node has {self.available_cpu_nums}")
        with self._lock:
            free = sorted(
                (int(k) for k, v in self.cpu_status_map.items() if v == 0)
            )
            if len(free) < num:
                raise CpuNotEnough(f"requested {num}, only {len(free)} free")
            if preferred_nodes and self.cpu_nodes:
                pref = set(preferred_nodes)
                local = [c for c in free if self.cpu_nodes.get(c) in pref]
                rest = [c for c in free if self.cpu_nodes.get(c) not in pref]
                chosen = (local + rest)[:num]
                chosen.sort()
            else:
                chosen = free[:num]
            for c in chosen:
                self.cpu_status_map[str(c)] = 1
            self._persist_async()
            return ",".join(str(c) for c in chosen)

    def apply_specific(self, cpuset: str | List[str]) -> None:
        """Re-acquire an exact cpuset (startup of a stopped container)."""
        ids = cpuset.split(",") if isinstance(cpuset, str) else list(cpuset)
        ids = [i.strip() for i in ids if i and i.strip()]
        if not ids:
            return
        with self._lock:
            busy = [i for i in ids if self.cpu_status_map.get(i, 1) == 1]
            if busy:
                raise CpuNotEnough(f"CPUs already allocated: {busy}")
            for i in ids:
                self.cpu_status_map[i] = 1
            self._persist_async()

    def restore(self, cpuset: str | List[str]) -> None:
        ids = cpuset.split(",") if isinstance(cpuset, str) else list(cpuset)
        ids = [i.strip() for i in ids if i and i.strip()]
        if not ids:
            return
        with self._lock:
            for i in ids:
                if i in self.cpu_status_map:
                    self.cpu_status_map[i] = 0
            self._persist_async()

    def get_cpu_status(self) -> Dict[str, int]:
        with self._lock:
            return dict(self.cpu_status_map)
