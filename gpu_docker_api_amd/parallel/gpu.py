"""GPU scheduler: xGMI-topology-aware bitmap allocator.

Contract parity with the reference (gpuscheduler.go:85-157): Apply(n) marks n
free GPUs used and returns their IDs, Restore flips them back, status is the
uuid -> 0|1 map, and the whole state round-trips through the store under
``gpus/gpuStatusMapKey`` with the same JSON shape
(``{"availableGpuNums": N, "gpuStatusMap": {...}}``).

Differences by design:
* placement is topology-aware (Topology.best_subset) instead of first-fit
  over Go's random map iteration;
* per-GPU HBM usage and the xGMI adjacency matrix are exposed for the
  Resource API (the reference reports only a used/free bit);
* allocation can exclude GPUs whose free HBM is below a requested floor.
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional

from ..state.keys import GPU_STATUS_MAP_KEY, Resource
from ..state.store import StateStore
from ..state.workqueue import WorkQueue
from ..xerrors import GpuNotEnough
from .base import BaseScheduler
from .inventory import GpuInfo, GpuInventory
from .topology import Topology


class GpuScheduler(BaseScheduler):
    resource = Resource.GPUS
    state_key = GPU_STATUS_MAP_KEY

    def __init__(
        self,
        store: StateStore,
        queue: Optional[WorkQueue],
        inventory: GpuInventory,
    ) -> None:
        super().__init__(store, queue)
        self.inventory = inventory
        self.gpus: List[GpuInfo] = []
        # total GPUs on the node — NOT a free count. The reference persists
        # this as "availableGpuNums" with the same never-decremented meaning
        # (gpuscheduler.go:32-37); the wire field name is kept for state
        # compatibility, the Python name says what it is.
        # The real free-count check happens under the lock in apply().
        self.node_gpu_count = 0
        self.gpu_status_map: Dict[str, int] = {}
        self.topology: Topology = Topology([], [])

    @classmethod
    async def create(
        cls,
        store: StateStore,
        queue: Optional[WorkQueue],
        inventory: GpuInventory,
        probe: Optional[dict] = None,
    ) -> "GpuScheduler":
        self = cls(store, queue, inventory)
        self.gpus = inventory.enumerate()
        uuids = [g.uuid for g in self.gpus]
        self.topology = Topology(inventory.link_matrix(), uuids)
        if probe:
            self.topology.overlay_measured(probe)
        kv = await store.get_or_none(self.resource, self.state_key)
        if kv is not None:
            data = json.loads(kv.value)
            persisted = data.get("gpuStatusMap") or {}
            # keep persisted bits for GPUs that still exist; new GPUs are free
            self.gpu_status_map = {u: int(persisted.get(u, 0)) for u in uuids}
            self.node_gpu_count = len(uuids)
        else:
            self.gpu_status_map = {u: 0 for u in uuids}
            self.node_gpu_count = len(uuids)
            await self.persist()
        return self

    # ------------------------------------------------------------------ api
    def serialize(self) -> str:
        with self._lock:
            return json.dumps(
                {
                    "availableGpuNums": self.node_gpu_count,
                    "gpuStatusMap": self.gpu_status_map,
                },
                separators=(",", ":"),
            )

    def apply(self, num: int, min_free_hbm: int = 0) -> List[str]:
        """Allocate ``num`` GPUs, topology-aware. Returns their UUIDs."""
        with self._lock:
            if num <= 0 or num > self.node_gpu_count:
                raise GpuNotEnough(
                    f"requested {num}, node has {self.node_gpu_count}"
                )
            free_idx = [
                g.index
                for g in self.gpus
                if self.gpu_status_map.get(g.uuid, 1) == 0
                and (min_free_hbm <= 0 or (g.vram_total - g.vram_used) >= min_free_hbm)
            ]
            if len(free_idx) < num:
                raise GpuNotEnough(f"requested {num}, only {len(free_idx)} free")
            chosen = self.topology.best_subset(free_idx, num)
            uuids = [self.gpus[i].uuid for i in chosen]
            for u in uuids:
                self.gpu_status_map[u] = 1
            self._persist_async()
            return uuids

    def apply_specific(self, uuids: List[str]) -> None:
        """Re-acquire an exact GPU set (startup of a stopped container,
        which released its resources — reference just restarts without
        re-acquiring, letting a stopped container's GPUs be double-booked).
        Raises GpuNotEnough if any is already allocated."""
        if not uuids:
            return
        with self._lock:
            busy = [u for u in uuids if self.gpu_status_map.get(u, 1) == 1]
            if busy:
                raise GpuNotEnough(f"GPUs already allocated: {busy}")
            unknown = [u for u in uuids if u not in self.gpu_status_map]
            if unknown:
                raise GpuNotEnough(f"unknown GPUs: {unknown}")
            for u in uuids:
                self.gpu_status_map[u] = 1
            self._persist_async()

    def restore(self, uuids: List[str]) -> None:
        if not uuids:
            return
        with self._lock:
            for u in uuids:
                if u in self.gpu_status_map:
                    self.gpu_status_map[u] = 0
            self._persist_async()

    def get_gpu_status(self) -> Dict[str, int]:
        with self._lock:
            return dict(self.gpu_status_map)

    def get_detail(self) -> dict:
        """Rich status for the Resource API: per-GPU HBM + adjacency."""
        with self._lock:
            self.gpus = self.inventory.refresh_usage() or self.gpus
            return {
                "availableGpuNums": self.node_gpu_count,
                "gpuStatusMap": dict(self.gpu_status_map),
                "gpus": [
                    {**g.to_dict(), "allocated": bool(self.gpu_status_map.get(g.uuid, 0))}
                    for g in self.gpus
                ],
                "xgmi": self.topology.to_dict(),
            }

    def info_by_uuid(self, uuid: str) -> Optional[GpuInfo]:
        for g in self.gpus:
            if g.uuid == uuid:
                return g
        return None
