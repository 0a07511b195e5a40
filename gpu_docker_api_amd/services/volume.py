"""Volume service: versioned volumes with data-preserving resize.

Parity with the reference (/root/reference/internal/services/volume.go:24-247)
with its bugs fixed:

* size comparison is by *bytes*, not string equality ("20GB" == "20480MB";
  the reference's string compare, volume.go:116-122, calls that a change);
* the migration move is awaited (the reference's helper-container ExecStart
  is fire-and-forget, utils/copy.go:122-124) and runs host-side through the
  CopyEngine — no throwaway ubuntu container;
* the old volume IS deleted after a successful resize (the reference leaks
  it — volume.go:155-159 commented out);
* shrink guard: refuse when used bytes exceed the new size
  (volume.go:126-140 semantics kept).
"""
from __future__ import annotations

import asyncio

import logging
from typing import Dict, List, Optional

from .saga import Saga
from ..config import Config
from ..models import VolumeCreate
from ..models.etcd import HistoryItem, VolumeSpec, _now_str
from ..models.memory import to_bytes
from ..runtime.base import RuntimeDriver
from ..state.keys import Resource
from ..state.store import StateStore
from ..state.workqueue import WorkQueue
from ..utils.copy import CopyEngine
from ..utils.files import dir_size
from ..utils.timing import PhaseTimer
from ..version import VersionMap
from ..xerrors import (
    NoPatchRequired,
    NotExistInStore,
    VolumeExisted,
    VolumeSizeUsedGreaterThanReduced,
)

log = logging.getLogger(__name__)


def versioned(name: str, version: int) -> str:
    return f"{name}-{version}"


class VolumeService:
    def __init__(
        self,
        *,
        store: StateStore,
        queue: Optional[WorkQueue],
        versions: VersionMap,
        runtime: RuntimeDriver,
        copy_engine: Optional[CopyEngine] = None,
        cfg: Optional[Config] = None,
    ) -> None:
        self.store = store
        self.queue = queue
        self.versions = versions
        self.runtime = runtime
        self.copy = copy_engine or CopyEngine()
        self.cfg = cfg or Config()

    async def _persist(self, name: str, spec: VolumeSpec) -> None:
        # synchronous: volume specs are read back by resize/history (see
        # ReplicaSetService._persist_spec for the rationale)
        await self.store.put(Resource.VOLUMES, name, spec.serialize())

    async def _load_spec(self, name: str) -> VolumeSpec:
        kv = await self.store.get_or_none(Resource.VOLUMES, name)
        if kv is None:
            raise NotExistInStore(name)
        return VolumeSpec.deserialize(kv.value)

    # ----------------------------------------------------------------- create
    async def create_volume(self, req: VolumeCreate) -> Dict:
        """POST /api/v1/volumes (reference CreateVolume, volume.go:24-52)."""
        timer = PhaseTimer("volume.create")
        name = req.name
        if self.versions.exists(name) or await self.runtime.volume_inspect(name):
            raise VolumeExisted(name)
        if req.size:
            to_bytes(req.size)  # validate unit
        async with Saga("volume.create") as saga:
            spec, vname = await self._create_versioned(saga, name, req.size)
            saga.commit()
        phases = timer.finish()
        return {"name": vname, "size": spec.size, "phases": phases}

    async def _create_versioned(self, saga: Saga, name: str, size: str):
        old_version = self.versions.get(name)
        version = self.versions.bump(name)
        saga.push(
            "version",
            lambda: self.versions.set(name, old_version)
            if old_version is not None
            else self.versions.remove(name),
        )
        vname = versioned(name, version)
        driver_opts = {"size": size} if size else {}
        vs = await self.runtime.volume_create(vname, driver_opts or None)
        saga.push("volume", lambda: self.runtime.volume_remove(vname, force=True))
        spec = VolumeSpec(
            version=version,
            create_time=_now_str(),
            opt={
                "Name": vname,
                "Driver": vs.driver,
                "DriverOpts": driver_opts,
            },
        )
        # compensate the spec write too: a failed later step (migration)
        # must not leave a v+1 spec disagreeing with the reverted version
        # map (same class of gap ADVICE r1 #2 found on the container side)
        prev = await self.store.get_or_none(Resource.VOLUMES, name)

        async def _undo_persist(prev=prev):
            if prev is None:
                await self.store.delete(Resource.VOLUMES, name)
            else:
                await self.store.put(Resource.VOLUMES, name, prev.value)

        await self._persist(name, spec)
        saga.push("spec", _undo_persist)
        return spec, vname

    # ----------------------------------------------------------------- resize
    async def patch_volume_size(self, name: str, new_size: str) -> Dict:
        """PATCH /api/v1/volumes/{name}/size (reference PatchVolumeSize,
        volume.go:98-176): create volume <name>-<v+1> with the new size and
        migrate the data."""
        timer = PhaseTimer("volume.patch")
        spec = await self._load_spec(name)
        cur_version = self.versions.get(name)
        if cur_version is None:
            raise NotExistInStore(name)
        vname_old = versioned(name, cur_version)
        new_bytes = to_bytes(new_size)
        old_bytes = to_bytes(spec.size) if spec.size else 0
        if old_bytes and new_bytes == old_bytes:
            raise NoPatchRequired(f"{name}: same size")
        old_vs = await self.runtime.volume_inspect(vname_old)
        if new_bytes < old_bytes and old_vs is not None:
            # fs walk of a large volume must not stall the event loop
            used = await asyncio.get_running_loop().run_in_executor(
                None, dir_size, old_vs.mountpoint
            )
            if used > new_bytes:
                raise VolumeSizeUsedGreaterThanReduced(
                    f"{name}: used {used} > requested {new_bytes}"
                )
        timer.mark("validate")
        async with Saga("volume.patch") as saga:
            new_spec, vname_new = await self._create_versioned(saga, name, new_size)
            timer.mark("create")
            new_vs = await self.runtime.volume_inspect(vname_new)
            if old_vs is not None and new_vs is not None:
                await self.copy.move_contents(old_vs.mountpoint, new_vs.mountpoint)
            timer.mark("migrate")
            saga.commit()
        # delete the old versioned volume (the reference leaks it)
        try:
            await self.runtime.volume_remove(vname_old, force=True)
        except Exception:
            log.exception("removing old volume %s failed", vname_old)
        phases = timer.finish()
        return {"name": vname_new, "size": new_spec.size, "phases": phases}

    # ----------------------------------------------------------------- delete
    async def delete_volume(self, name: str, keep_record: bool = False) -> None:
        """DELETE /api/v1/volumes/{name}[?noall=1] (reference DeleteVolume,
        volume.go:178-219): remove the runtime volume; unless keep_record,
        also drop the store record + version entry."""
        version = self.versions.get(name)
        if version is None:
            raise NotExistInStore(name)
        vname = versioned(name, version)
        try:
            await self.runtime.volume_remove(vname, force=True)
        except Exception:
            log.exception("volume remove %s failed", vname)
        if not keep_record:
            self.versions.remove(name)
            await self.store.delete(Resource.VOLUMES, name)

    # ------------------------------------------------------------------- info
    async def list_volumes(self) -> List[Dict]:
        """All volumes with spec + live state (extension)."""
        out = []
        for name, version in sorted(self.versions.snapshot().items()):
            kv = await self.store.get_or_none(Resource.VOLUMES, name)
            if kv is None:
                continue
            spec = VolumeSpec.deserialize(kv.value)
            vs = await self.runtime.volume_inspect(spec.name)
            out.append(
                {
                    "name": name,
                    "volumeName": spec.name,
                    "version": version,
                    "size": spec.size,
                    "mountpoint": vs.mountpoint if vs else "",
                    "present": vs is not None,
                }
            )
        return out

    async def get_volume_info(self, name: str) -> Dict:
        spec = await self._load_spec(name)
        out = spec.to_dict()
        # surface live quota enforcement: "loop" (ENOSPC-enforced) vs
        # "none" (advisory), plus any degradation recorded by the runtime
        # (e.g. a failed remount after daemon restart) — a silent downgrade
        # would misrepresent the size guarantee
        vs = await self.runtime.volume_inspect(spec.name)
        if vs is not None:
            opts = vs.options or {}
            if "enforced" in opts:
                out["sizeEnforced"] = opts.get("enforced")
            if opts.get("degraded"):
                out["degraded"] = opts.get("degraded")
        return out

    async def get_volume_history(self, name: str) -> List[Dict]:
        kvs = await self.store.history(Resource.VOLUMES, name)
        out = []
        for kv in kvs:
            spec = VolumeSpec.deserialize(kv.value)
            out.append(
                HistoryItem(
                    version=kv.version, create_time=spec.create_time, status=spec.to_dict()
                ).to_dict()
            )
        return out
