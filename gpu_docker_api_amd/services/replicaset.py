"""ReplicaSet service: versioned container lifecycle with rolling replacement.

Functional parity with the reference's ReplicaSetService
(/root/reference/internal/services/replicaset.go:45-1047 and
replicaset_nomock.go:25-140), redesigned:

* every multi-step flow is a :class:`Saga` with complete compensation (the
  reference's manual unwind is incomplete — SURVEY.md §7.3 item 2);
* GPU attachment is ROCm device/visibility injection chosen by the runtime
  driver, never an NVIDIA runtime (replicaset_nomock.go:128-140);
* the replaced version's writable layer is actually preserved under
  ``merges/<name>/<name>-<version>`` (the reference's preservation copy is
  commented out — replicaset.go:688-698 — leaving only an empty dir);
* a stopped container's resources are re-acquired exactly (apply_specific)
  at startup — the reference restarts without re-acquiring, allowing
  double-booking of its released GPUs;
* the "ballast stone" hack (a 5 MB dd file created 5 s after start,
  replicaset.go:1013-1047, to keep overlay2 quota metadata warm) is
  dropped: quota behavior belongs to the storage driver, and none of our
  three runtime drivers needs it. Documented here for the record.

Operation phases are timed (utils.timing) because the headline metric is
create->running / patch turnaround latency.
"""
from __future__ import annotations

import asyncio
import logging
import os
import shutil
from typing import Dict, List, Optional, Tuple

from ..config import Config
from ..models import ContainerCommit, ContainerExecute, ContainerRun, PatchRequest
from ..models.etcd import ContainerSpec, HistoryItem, _now_str
from ..models.memory import to_bytes
from ..parallel import CpuScheduler, GpuScheduler, PortScheduler
from ..runtime.base import ContainerState, RuntimeDriver
from ..state.keys import Resource
from ..state.store import StateStore
from ..state.workqueue import WorkQueue
from ..utils.copy import CopyEngine
from ..utils.names import safe_subpath
from ..utils.timing import PhaseTimer
from ..state.keys import RELEASED_SET_KEY
from ..version import MergeMap, ReleasedSet, VersionMap
from ..xerrors import (
    ContainerExisted,
    ContainerNotExist,
    NoPatchRequired,
    NoRollbackRequired,
)
from .saga import Saga

log = logging.getLogger(__name__)


def versioned(name: str, version: int) -> str:
    return f"{name}-{version}"


class ReplicaSetService:
    def __init__(
        self,
        *,
        store: StateStore,
        queue: Optional[WorkQueue],
        gpu: GpuScheduler,
        cpu: CpuScheduler,
        ports: PortScheduler,
        versions: VersionMap,
        merges: MergeMap,
        runtime: RuntimeDriver,
        copy_engine: Optional[CopyEngine] = None,
        cfg: Optional[Config] = None,
        released: Optional[ReleasedSet] = None,
    ) -> None:
        self.store = store
        self.queue = queue
        self.gpu = gpu
        self.cpu = cpu
        self.ports = ports
        self.versions = versions
        self.merges = merges
        self.runtime = runtime
        self.copy = copy_engine or CopyEngine()
        self.cfg = cfg or Config()
        # versioned names whose resources were released by stop_container —
        # PERSISTED (write-behind) so a daemon restart cannot forget a stop
        # and double-book the freed GPUs; Daemon.start
        # loads it before serving
        self._released: ReleasedSet = released or ReleasedSet(
            store, queue, RELEASED_SET_KEY
        )

    # ------------------------------------------------------------ persistence
    async def _persist_spec(self, name: str, spec: ContainerSpec) -> None:
        """Container specs are the read-path source of truth (history,
        patch, rollback all read them back), so they are written
        synchronously; only scheduler/version-map state is write-behind.
        The reference queues spec writes too (replicaset.go:149-153),
        leaving a window where a patch can read a stale spec."""
        await self.store.put(Resource.CONTAINERS, name, spec.serialize())

    async def _load_spec(self, name: str) -> ContainerSpec:
        kv = await self.store.get_or_none(Resource.CONTAINERS, name)
        if kv is None:
            raise ContainerNotExist(name)
        return ContainerSpec.deserialize(kv.value)

    def _current_vname(self, name: str) -> str:
        v = self.versions.get(name)
        if v is None:
            raise ContainerNotExist(name)
        return versioned(name, v)

    async def _spec_at_version(self, name: str, version: int) -> ContainerSpec:
        """The stored spec of history version N, matched on the spec's OWN
        version field rather than the store's per-key version counter.
        The counters usually align (one put per container version), but a
        compensated saga re-put or an etcd backend shared with other
        writers advances the per-key counter without a container-version
        bump — matching on content is robust to that."""
        for kv in await self.store.history(Resource.CONTAINERS, name):
            spec = ContainerSpec.deserialize(kv.value)
            if spec.version == version:
                return spec
        raise ContainerNotExist(f"{name} version={version}")

    # ----------------------------------------------------------------- create
    async def run_gpu_container(self, req: ContainerRun) -> Dict:
        """POST /api/v1/replicaSet (reference RunGpuContainer,
        replicaset.go:45-155). Create + start, returns the versioned name."""
        name = req.replica_set_name
        timer = PhaseTimer("replicaset.run")
        if self.versions.exists(name) or await self._exist_container_any(name):
            raise ContainerExisted(name)

        spec = ContainerSpec()
        spec.config = {
            "Image": req.image_name,
            "Env": list(req.env),
            "Cmd": list(req.cmd),
            "Labels": {"gda.replicaSet": name},
        }
        spec.host_config = {
            "Binds": [b.format() for b in req.binds if b.format()],
            "RestartPolicy": {"Name": self.cfg.restart_policy},
            "ShmSize": self.cfg.shm_size_bytes,
            # rootfs quota (requires overlay2-on-xfs on the docker driver;
            # reference hardcodes 30G: replicaset.go:67-69)
            "StorageOpt": {"size": self.cfg.rootfs_quota},
        }
        if req.container_ports:
            spec.config["Labels"]["gda.containerPorts"] = ",".join(req.container_ports)

        async with Saga("run") as saga:
            if req.gpu_count > 0:
                min_hbm = to_bytes(req.gpu_memory) if req.gpu_memory else 0
                uuids = self.gpu.apply(req.gpu_count, min_free_hbm=min_hbm)
                saga.push("gpu", lambda: self.gpu.restore(uuids))
                spec.gpu_uuids = uuids
            else:
                spec.gpu_uuids = []
            if req.cpu_count > 0:
                nodes = self._gpu_numa_nodes(spec.gpu_uuids)
                cpuset = self.cpu.apply(req.cpu_count, preferred_nodes=nodes)
                saga.push("cpu", lambda: self.cpu.restore(cpuset))
                spec.cpuset_cpus = cpuset
                if nodes:
                    # memory locality too (docker HostConfig.CpusetMems; the
                    # proc runtime writes cgroup cpuset.mems)
                    spec.host_config["CpusetMems"] = ",".join(map(str, nodes))
            if req.memory:
                spec.memory_bytes = to_bytes(req.memory)
            timer.mark("schedule")
            cid, vname = await self._run_container(saga, name, spec, only_create=False, timer=timer)
            saga.commit()
        phases = timer.finish()
        return {"name": vname, "id": cid, "phases": phases}

    def _gpu_numa_nodes(self, uuids: List[str]) -> List[int]:
        """NUMA nodes of the allocated GPUs (for socket-local cpusets)."""
        nodes = []
        for u in uuids or []:
            info = self.gpu.info_by_uuid(u)
            if info is not None and info.numa_node >= 0:
                nodes.append(info.numa_node)
        return sorted(set(nodes))

    async def _exist_container_any(self, name: str) -> bool:
        """True if any versioned container of this replicaSet exists (the
        reference checks only *running* ones — a bug, replicaset.go:949-959)."""
        for st in await self.runtime.list(all=True):
            if st.name.rsplit("-", 1)[0] == name:
                return True
        return False

    async def _run_container(
        self,
        saga: Saga,
        name: str,
        spec: ContainerSpec,
        only_create: bool,
        timer: Optional[PhaseTimer] = None,
    ) -> Tuple[str, str]:
        """Version bump + port allocation + create (+start) + persist —
        the reference's runContainer (replicaset_nomock.go:25-114)."""
        old_version = self.versions.get(name)
        version = self.versions.bump(name)
        saga.push(
            "version",
            lambda: self.versions.set(name, old_version)
            if old_version is not None
            else self.versions.remove(name),
        )
        vname = versioned(name, version)
        spec.version = version
        spec.container_name = vname
        spec.create_time = _now_str()
        spec.set_env("CONTAINER_VERSION", str(version))

        ports_label = (spec.config.get("Labels") or {}).get("gda.containerPorts", "")
        container_ports = [p for p in ports_label.split(",") if p]
        if container_ports:
            host_ports = self.ports.apply(len(container_ports))
            saga.push("ports", lambda: self.ports.restore(host_ports))
            exposed, bindings = {}, {}
            for cport, hport in zip(container_ports, host_ports):
                key = cport if "/" in cport else f"{cport}/tcp"
                exposed[key] = {}
                bindings[key] = [{"HostIp": "0.0.0.0", "HostPort": str(hport)}]
            spec.config["ExposedPorts"] = exposed
            spec.host_config["PortBindings"] = bindings

        cid = await self.runtime.create(spec)
        saga.push("container", lambda: self.runtime.remove(vname, force=True))
        if timer:
            timer.mark("create")
        if not only_create:
            await self.runtime.start(vname)
            if timer:
                timer.mark("start")
        # capture the previous spec so a later saga failure (copy/start of
        # the replacement, old-container removal) restores the CONTAINERS
        # key too — without this the store would keep the new spec while the
        # version map reverted, so _load_spec and _current_vname disagree
        # afterwards
        prev = await self.store.get_or_none(Resource.CONTAINERS, name)

        async def _undo_persist(prev=prev):
            if prev is None:
                await self.store.delete(Resource.CONTAINERS, name)
            else:
                await self.store.put(Resource.CONTAINERS, name, prev.value)

        await self._persist_spec(name, spec)
        saga.push("spec", _undo_persist)
        if timer:
            timer.mark("persist")
        return cid, vname

    # ------------------------------------------------------------------ patch
    async def patch_container(self, name: str, req: PatchRequest) -> Dict:
        """PATCH /api/v1/replicaSet/{name} — rolling replace
        (reference PatchContainer, replicaset.go:267-363)."""
        timer = PhaseTimer("replicaset.patch")
        spec = await self._load_spec(name)
        vname_old = self._current_vname(name)
        changed = False

        async with Saga("patch") as saga:
            if vname_old in self._released:
                # stopped container: its resources were released at stop —
                # re-acquire them first or the replacement double-books GPUs
                self._reacquire_released(saga, spec, vname_old)
            if req.gpu_patch is not None:
                changed |= self._patch_gpu(saga, spec, req.gpu_patch.gpu_count)
            if req.cpu_patch is not None:
                changed |= self._patch_cpu(saga, spec, req.cpu_patch.cpu_count)
            if req.memory_patch is not None and req.memory_patch.memory:
                new_bytes = to_bytes(req.memory_patch.memory)
                if new_bytes != spec.memory_bytes:
                    spec.memory_bytes = new_bytes
                    changed = True
            if req.volume_patch is not None:
                changed |= self._patch_volume(spec, req.volume_patch)
            if not req.empty() and not changed:
                raise NoPatchRequired(name)
            timer.mark("schedule")
            vname_new, cid = await self._replace(saga, name, spec, vname_old, timer)
            saga.commit()
        phases = timer.finish()
        return {"containerName": vname_new, "id": cid, "phases": phases}

    def _reacquire_released(self, saga: Saga, spec: ContainerSpec, vname: str) -> None:
        """Re-acquire the exact resources a stop released (raises when any
        is now taken by another replicaSet)."""
        if spec.gpu_uuids:
            self.gpu.apply_specific(spec.gpu_uuids)
            saga.push("gpu-reacquired", lambda: self.gpu.restore(spec.gpu_uuids))
        if spec.cpuset_cpus:
            self.cpu.apply_specific(spec.cpuset_cpus)
            saga.push("cpu-reacquired", lambda: self.cpu.restore(spec.cpuset_cpus))
        self._released.discard(vname)
        saga.push("mark-released", lambda: self._released.add(vname))

    def _patch_gpu(self, saga: Saga, spec: ContainerSpec, new_count: int) -> bool:
        old_uuids = spec.gpu_uuids
        if new_count == len(old_uuids):
            return False
        if old_uuids:
            self.gpu.restore(old_uuids)
            saga.push("gpu-reacquire-old", lambda: self.gpu.apply_specific(old_uuids))
        if new_count > 0:
            new_uuids = self.gpu.apply(new_count)
            saga.push("gpu-release-new", lambda: self.gpu.restore(new_uuids))
        else:
            new_uuids = []
        spec.gpu_uuids = new_uuids
        return True

    def _patch_cpu(self, saga: Saga, spec: ContainerSpec, new_count: int) -> bool:
        old_set = [c for c in spec.cpuset_cpus.split(",") if c]
        if new_count == len(old_set):
            return False
        if old_set:
            self.cpu.restore(old_set)
            saga.push("cpu-reacquire-old", lambda: self.cpu.apply_specific(old_set))
        new_cpuset = (
            self.cpu.apply(new_count, preferred_nodes=self._gpu_numa_nodes(spec.gpu_uuids))
            if new_count > 0
            else ""
        )
        if new_cpuset:
            saga.push("cpu-release-new", lambda: self.cpu.restore(new_cpuset))
        spec.cpuset_cpus = new_cpuset
        return True

    @staticmethod
    def _patch_volume(spec: ContainerSpec, vp) -> bool:
        if vp.old_bind is None or vp.new_bind is None:
            return False
        old_s, new_s = vp.old_bind.format(), vp.new_bind.format()
        if not old_s or not new_s:
            return False
        binds = spec.binds
        if old_s not in binds:
            return False
        spec.binds = [new_s if b == old_s else b for b in binds]
        return True

    async def _replace(
        self,
        saga: Saga,
        name: str,
        spec: ContainerSpec,
        vname_old: str,
        timer: PhaseTimer,
        migrate_src: Optional[str] = None,
    ) -> Tuple[str, str]:
        """The rolling-replace tail shared by patch/rollback/restart
        (reference replicaset.go:318-359): create new version, migrate the
        writable layer (or ``migrate_src`` — rollback --restore-data uses a
        preserved historical layer), start, preserve old layer, delete old."""
        old_state = await self.runtime.inspect(vname_old)
        cid, vname_new = await self._run_container(saga, name, spec, only_create=True, timer=timer)
        new_state = await self.runtime.inspect(vname_new)
        src = migrate_src or (old_state.upper_dir if old_state is not None else "")
        if (
            new_state is not None
            and src
            and new_state.upper_dir
            and os.path.isdir(src)
        ):
            await self.copy.copy_dir(src, new_state.upper_dir)
        timer.mark("copy")
        await self.runtime.start(vname_new)
        timer.mark("start")
        # preserve the replaced version's layer for history/rollback.
        # When the runtime owns the rootfs dir (proc/mock) a rename is enough
        # — the old container is deleted next anyway; docker's UpperDir
        # belongs to overlayfs, so it is copied.
        if old_state is not None and old_state.upper_dir and os.path.isdir(old_state.upper_dir):
            # safe_subpath: names are router-validated, but never trust a
            # join that feeds an rmtree (defense in depth — ADVICE r1 #1)
            merge_path = safe_subpath(self.cfg.merges_dir, name, vname_old)
            try:
                if getattr(self.runtime, "owns_rootfs", False):
                    os.makedirs(os.path.dirname(merge_path), exist_ok=True)
                    shutil.rmtree(merge_path, ignore_errors=True)
                    try:
                        os.rename(old_state.upper_dir, merge_path)
                    except OSError:  # cross-device: fall back to copying
                        os.makedirs(merge_path, exist_ok=True)
                        await self.copy.copy_dir(old_state.upper_dir, merge_path)
                else:
                    os.makedirs(merge_path, exist_ok=True)
                    await self.copy.copy_dir(old_state.upper_dir, merge_path)
                self.merges.set(vname_old, merge_path)
                self._prune_merge_layers(name)
            except Exception:
                log.exception("preserving %s layer failed", vname_old)
        timer.mark("preserve")
        # delete the old container, releasing its host ports
        if old_state is not None:
            self.ports.restore(self._host_ports(old_state))
            await self.runtime.remove(vname_old, force=True)
        self._released.discard(vname_old)
        timer.mark("delete_old")
        return vname_new, cid

    def _prune_merge_layers(self, name: str) -> None:
        """Retain only the newest cfg.keep_merge_layers preserved layers."""
        keep = self.cfg.keep_merge_layers
        if keep <= 0:
            return
        prefix = name + "-"
        entries = []
        for vname, path in self.merges.snapshot().items():
            if vname.startswith(prefix):
                try:
                    entries.append((int(vname[len(prefix):]), vname, path))
                except ValueError:
                    continue
        entries.sort(reverse=True)
        for _v, vname, path in entries[keep:]:
            self.merges.remove(vname)
            shutil.rmtree(path, ignore_errors=True)

    @staticmethod
    def _host_ports(state: ContainerState) -> List[int]:
        out: List[int] = []
        for _cport, bindings in (state.port_bindings or {}).items():
            for b in bindings or []:
                try:
                    out.append(int(b.get("HostPort", "0")))
                except (TypeError, ValueError):
                    continue
        return [p for p in out if p]

    # --------------------------------------------------------------- rollback
    async def rollback_container(
        self, name: str, target_version: int, restore_data: bool = False
    ) -> Dict:
        """PATCH /{name}/rollback (reference RollbackContainer,
        replicaset.go:365-446): restore the spec of history version N, with
        resources re-resolved against live state. Note: the reference's
        memory restore divides by 1024^2 and labels it GB — a 1024x
        inflation (replicaset.go:408) — we restore exact bytes.

        ``restore_data`` (extension): also restore that version's preserved
        writable layer from merges/ instead of migrating the current data —
        possible here because preservation is real (the reference's is a
        commented-out no-op, replicaset.go:688-698)."""
        timer = PhaseTimer("replicaset.rollback")
        cur_version = self.versions.get(name)
        if cur_version is None:
            raise ContainerNotExist(name)
        if cur_version == target_version:
            raise NoRollbackRequired(name)
        target = await self._spec_at_version(name, target_version)
        live = await self._load_spec(name)
        vname_old = versioned(name, cur_version)

        async with Saga("rollback") as saga:
            if vname_old in self._released:
                self._reacquire_released(saga, live, vname_old)
            # re-resolve GPU/CPU against live allocation state
            self._patch_gpu(saga, live, len(target.gpu_uuids))
            self._patch_cpu(saga, live, len([c for c in target.cpuset_cpus.split(",") if c]))
            # adopt target's config with live resource assignments
            new_spec = ContainerSpec.from_dict(target.to_dict())
            new_spec.gpu_uuids = live.gpu_uuids
            new_spec.cpuset_cpus = live.cpuset_cpus
            new_spec.memory_bytes = target.memory_bytes
            timer.mark("schedule")
            migrate_src = None
            if restore_data:
                preserved = self.merges.get(versioned(name, target_version))
                if not preserved or not os.path.isdir(preserved):
                    raise NoRollbackRequired(
                        f"{name}: no preserved layer for version {target_version} "
                        f"(retention keep_merge_layers={self.cfg.keep_merge_layers})"
                    )
                migrate_src = preserved
            vname_new, cid = await self._replace(
                saga, name, new_spec, vname_old, timer, migrate_src=migrate_src
            )
            saga.commit()
        phases = timer.finish()
        return {"containerName": vname_new, "id": cid, "phases": phases}

    # ------------------------------------------------------- stop/start/pause
    async def stop_container(self, name: str) -> None:
        """Stop + release GPU/CPU/ports (reference StopContainer,
        replicaset.go:582-639)."""
        spec = await self._load_spec(name)
        vname = self._current_vname(name)
        if vname in self._released:
            await self.runtime.stop(vname)
            return
        self.gpu.restore(spec.gpu_uuids)
        self.cpu.restore(spec.cpuset_cpus)
        st = await self.runtime.inspect(vname)
        if st is not None:
            self.ports.restore(self._host_ports(st))
        self._released.add(vname)
        await self.runtime.stop(vname)

    async def startup_container(self, name: str) -> None:
        """'continue' (reference StartupContainer, replicaset.go:717-734):
        restart the current version. Unlike the reference we re-acquire the
        exact resources a stop released before starting."""
        spec = await self._load_spec(name)
        vname = self._current_vname(name)
        if vname in self._released:
            async with Saga("startup") as saga:
                if spec.gpu_uuids:
                    self.gpu.apply_specific(spec.gpu_uuids)
                    saga.push("gpu", lambda: self.gpu.restore(spec.gpu_uuids))
                if spec.cpuset_cpus:
                    self.cpu.apply_specific(spec.cpuset_cpus)
                    saga.push("cpu", lambda: self.cpu.restore(spec.cpuset_cpus))
                hports = [
                    int(b["HostPort"])
                    for bl in (spec.host_config.get("PortBindings") or {}).values()
                    for b in bl or []
                    if b.get("HostPort")
                ]
                if hports:
                    self.ports.apply_specific(hports)
                    saga.push("ports", lambda: self.ports.restore(hports))
                await self.runtime.restart(vname)
                saga.commit()
            self._released.discard(vname)
        else:
            await self.runtime.restart(vname)

    async def pause_container(self, name: str) -> None:
        await self.runtime.pause(self._current_vname(name))

    async def restart_container(self, name: str) -> Dict:
        """PATCH /{name}/restart — rolling replace re-applying resources
        (reference RestartContainer, replicaset.go:736-864)."""
        timer = PhaseTimer("replicaset.restart")
        spec = await self._load_spec(name)
        vname_old = self._current_vname(name)
        async with Saga("restart") as saga:
            if vname_old in self._released:
                # stopped: re-acquire fresh resources of the same shape
                if spec.gpu_uuids:
                    uuids = self.gpu.apply(len(spec.gpu_uuids))
                    saga.push("gpu", lambda: self.gpu.restore(uuids))
                    spec.gpu_uuids = uuids
                if spec.cpuset_cpus:
                    cpuset = self.cpu.apply(
                        len(spec.cpuset_cpus.split(",")),
                        preferred_nodes=self._gpu_numa_nodes(spec.gpu_uuids),
                    )
                    saga.push("cpu", lambda: self.cpu.restore(cpuset))
                    spec.cpuset_cpus = cpuset
                self._released.discard(vname_old)
            timer.mark("schedule")
            vname_new, cid = await self._replace(saga, name, spec, vname_old, timer)
            saga.commit()
        phases = timer.finish()
        return {"containerName": vname_new, "id": cid, "phases": phases}

    # ------------------------------------------------------------ delete/exec
    async def delete_container(self, name: str) -> None:
        """DELETE (reference DeleteContainer, replicaset.go:157-223)."""
        spec = await self._load_spec(name)
        vname = self._current_vname(name)
        if vname not in self._released:
            self.gpu.restore(spec.gpu_uuids)
            self.cpu.restore(spec.cpuset_cpus)
            st = await self.runtime.inspect(vname)
            if st is not None:
                self.ports.restore(self._host_ports(st))
        self._released.discard(vname)
        # wipe preserved layers + merge map entries + version map + store key
        merges_path = safe_subpath(self.cfg.merges_dir, name)
        await asyncio.get_running_loop().run_in_executor(
            None, lambda: shutil.rmtree(merges_path, ignore_errors=True)
        )
        self.merges.remove_prefix(name + "-")
        self.versions.remove(name)
        await self.store.delete(Resource.CONTAINERS, name)
        try:
            await self.runtime.remove(vname, force=True)
        except ContainerNotExist:
            pass

    async def execute_container(self, name: str, req: ContainerExecute):
        """POST /{name}/execute (reference ExecuteContainer,
        replicaset.go:225-265). Returns (stdout, exit code) — the exit code
        is an extension (the reference returns stdout only)."""
        vname = self._current_vname(name)
        return await self.runtime.execute_rc(vname, list(req.cmd), req.work_dir)

    async def commit_container(self, name: str, req: ContainerCommit) -> str:
        """POST /{name}/commit (reference CommitContainer,
        replicaset.go:866-894; its empty-name tag bug fixed by router
        validation)."""
        vname = self._current_vname(name)
        return await self.runtime.commit(vname, req.new_image_name)

    # ------------------------------------------------------------------ info
    async def list_replicasets(self) -> List[Dict]:
        """All replicaSets with their stored spec + live runtime state
        (extension: the reference has no list endpoint)."""
        out = []
        live = {st.name: st for st in await self.runtime.list(all=True)}
        for name, version in sorted(self.versions.snapshot().items()):
            kv = await self.store.get_or_none(Resource.CONTAINERS, name)
            if kv is None:
                continue
            spec = ContainerSpec.deserialize(kv.value)
            st = live.get(spec.container_name)
            out.append(
                {
                    "name": name,
                    "containerName": spec.container_name,
                    "version": version,
                    "image": spec.image,
                    "gpuCount": len(spec.gpu_uuids),
                    "cpuset": spec.cpuset_cpus,
                    "memory": spec.memory_bytes,
                    "status": st.status if st else "unknown",
                    "running": bool(st.running) if st else False,
                }
            )
        return out

    async def get_container_info(self, name: str) -> Dict:
        kv = await self.store.get_or_none(Resource.CONTAINERS, name)
        if kv is None:
            raise ContainerNotExist(name)
        return ContainerSpec.deserialize(kv.value).to_dict()

    async def get_container_logs(self, name: str, tail: int = 200) -> str:
        """GET /{name}/logs (extension): captured console output of the
        current version. The reference exposes no logs route — its users
        must query dockerd directly."""
        return await self.runtime.logs(self._current_vname(name), tail=tail)

    async def get_container_stats(self, name: str) -> Dict:
        """GET /{name}/stats (extension): live cpu/memory/pids of the
        current version, plus per-allocated-GPU HBM usage (288 GB HBM3E
        per MI355X; the reference reports no per-container usage at all)."""
        spec = await self._load_spec(name)
        out = await self.runtime.stats(self._current_vname(name))
        if spec.gpu_uuids:
            try:
                # amdsmi reads can take tens of ms: off the event loop
                gpus = await asyncio.get_running_loop().run_in_executor(
                    None, self.gpu.inventory.refresh_usage
                )
                fresh = {g.uuid: g for g in (gpus or [])}
            except Exception:  # inventory backend without live usage
                fresh = {}
            out["gpus"] = [
                {
                    "uuid": u,
                    "hbmUsedBytes": fresh[u].vram_used if u in fresh else None,
                    "hbmTotalBytes": fresh[u].vram_total if u in fresh else None,
                }
                for u in spec.gpu_uuids
            ]
        return out

    async def get_container_history(self, name: str) -> List[Dict]:
        """GET /{name}/history — all surviving versions, newest first
        (reference GetContainerHistory, replicaset.go:908-929, via the
        etcd revision walker)."""
        try:
            kvs = await self.store.history(Resource.CONTAINERS, name)
        except Exception as exc:
            raise ContainerNotExist(name) from exc
        out = []
        for kv in kvs:
            spec = ContainerSpec.deserialize(kv.value)
            out.append(
                HistoryItem(
                    version=kv.version, create_time=spec.create_time, status=spec.to_dict()
                ).to_dict()
            )
        return out
