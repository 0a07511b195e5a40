"""Daemon wiring (the analog of the reference's program.Init/Start/Stop,
/root/reference/cmd/gpu-docker-api/main.go:53-154).

Init order mirrors main.go: state store -> work queue -> GPU/CPU/port
schedulers -> version maps -> merges dir -> services -> HTTP routes.
Shutdown drains the work queue then persists all scheduler/version state
synchronously (main.go:139-154) — plus closes the runtime and the WAL.
"""
from __future__ import annotations

import asyncio
import logging
import os
from contextlib import asynccontextmanager
from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import PlainTextResponse

from ..config import Config
from ..parallel import CpuScheduler, GpuScheduler, PortScheduler, make_inventory
from ..parallel.topology import Topology
from ..runtime import make_runtime
from ..services import ReplicaSetService, VolumeService
from ..state.etcd_gateway import EtcdGatewayStore
from ..state.keys import (
    CONTAINER_MERGE_MAP_KEY,
    CONTAINER_VERSION_MAP_KEY,
    RELEASED_SET_KEY,
    VOLUME_VERSION_MAP_KEY,
)
from ..state.store import MemoryStore, StateStore
from ..state.workqueue import WorkQueue
from ..utils.copy import CopyEngine
from ..utils.timing import METRICS
from ..version import MergeMap, ReleasedSet, VersionMap
from . import replicaset as replicaset_router
from . import resource as resource_router
from . import volume as volume_router
from .middleware import AuthMiddleware, CorsMiddleware

log = logging.getLogger(__name__)


class Daemon:
    """Holds every initialized subsystem; built once per process."""

    def __init__(self, cfg: Config, store: Optional[StateStore] = None) -> None:
        self.cfg = cfg
        self.store: Optional[StateStore] = store
        self.queue: Optional[WorkQueue] = None
        self.gpu: Optional[GpuScheduler] = None
        self.cpu: Optional[CpuScheduler] = None
        self.ports: Optional[PortScheduler] = None
        self.container_versions: Optional[VersionMap] = None
        self.volume_versions: Optional[VersionMap] = None
        self.merges: Optional[MergeMap] = None
        self.runtime = None
        self.replicaset: Optional[ReplicaSetService] = None
        self.volume: Optional[VolumeService] = None

    async def start(self) -> None:
        cfg = self.cfg
        os.makedirs(cfg.data_dir, exist_ok=True)
        os.makedirs(cfg.merges_dir, exist_ok=True)

        if self.store is None:
            if cfg.state.startswith("etcd:"):
                self.store = EtcdGatewayStore(
                    cfg.state.split(":", 1)[1] or "http://127.0.0.1:2379"
                )
            elif cfg.state == "memory":
                self.store = MemoryStore(wal_path=cfg.wal_path)
            else:
                self.store = MemoryStore()
        self.queue = WorkQueue(self.store)
        self.queue.start()

        inventory = make_inventory(cfg.inventory, cfg.mock_gpus)
        probe = None
        if cfg.probe_cache:
            probe = Topology.load_probe_file(cfg.probe_cache)
        if probe is None and cfg.run_xgmi_probe:
            probe = await self._run_probe()
            if probe is not None and cfg.probe_cache:
                # cache the measured link map for the next startup
                try:
                    import json as _json

                    with open(cfg.probe_cache, "w") as f:
                        _json.dump(probe, f)
                except OSError as exc:
                    log.warning("could not write probe cache: %s", exc)
        self.gpu = await GpuScheduler.create(self.store, self.queue, inventory, probe=probe)
        self.cpu = await CpuScheduler.create(self.store, self.queue)
        self.ports = await PortScheduler.create(
            self.store, self.queue, cfg.port_start, cfg.port_end
        )

        self.container_versions = VersionMap(self.store, self.queue, CONTAINER_VERSION_MAP_KEY)
        self.volume_versions = VersionMap(self.store, self.queue, VOLUME_VERSION_MAP_KEY)
        self.merges = MergeMap(self.store, self.queue, CONTAINER_MERGE_MAP_KEY)
        self.released = ReleasedSet(self.store, self.queue, RELEASED_SET_KEY)
        await self.container_versions.load()
        await self.volume_versions.load()
        await self.merges.load()
        await self.released.load()

        rt_kwargs = {}
        if cfg.runtime in ("proc",):
            rt_kwargs = {
                "base_dir": os.path.join(cfg.data_dir, "procrt"),
                "gpu_resolver": self.gpu.info_by_uuid,
                "loop_volumes": cfg.loop_volumes,
            }
        elif cfg.runtime == "mock":
            rt_kwargs = {"base_dir": os.path.join(cfg.data_dir, "mockrt")}
        elif cfg.runtime == "docker":
            rt_kwargs = {
                "socket_path": cfg.docker_socket,
                "gpu_resolver": self.gpu.info_by_uuid,
                "use_cdi": cfg.use_cdi,
            }
        self.runtime = make_runtime(cfg.runtime, **rt_kwargs)

        copy_engine = CopyEngine(cfg.copy_engine)
        self.replicaset = ReplicaSetService(
            store=self.store,
            queue=self.queue,
            gpu=self.gpu,
            cpu=self.cpu,
            ports=self.ports,
            versions=self.container_versions,
            merges=self.merges,
            runtime=self.runtime,
            copy_engine=copy_engine,
            cfg=cfg,
            released=self.released,
        )
        self.volume = VolumeService(
            store=self.store,
            queue=self.queue,
            versions=self.volume_versions,
            runtime=self.runtime,
            copy_engine=copy_engine,
            cfg=cfg,
        )
        if cfg.run_rccl_smoke:
            await self._run_rccl_smoke()
        self._janitor = asyncio.get_running_loop().create_task(self._janitor_loop())

    async def _janitor_loop(self) -> None:
        """Periodic housekeeping: glibc keeps freed churn memory in arenas
        (observed ~5 KB/cycle RSS creep under sustained load on a 256-core
        box while the Python heap stays bounded per tracemalloc);
        malloc_trim returns it to the kernel."""
        import ctypes
        import gc

        try:
            libc = ctypes.CDLL("libc.so.6")
        except OSError:
            return
        while True:
            await asyncio.sleep(60.0)
            gc.collect()
            try:
                libc.malloc_trim(0)
            except Exception:
                return

    async def _run_probe(self) -> Optional[dict]:
        """Run the native xGMI bandwidth probe (csrc -> ops.hipcore)."""
        try:
            from ..ops import hipcore

            return await hipcore.run_probe_async()
        except Exception as exc:  # no GPU / extension not built
            log.warning("xGMI probe unavailable: %s", exc)
            return None

    async def _run_rccl_smoke(self) -> None:
        try:
            from ..ops import hipcore

            result = await hipcore.run_rccl_smoke_async()
            log.info("RCCL smoke: %s", result)
        except Exception as exc:
            log.warning("RCCL smoke unavailable: %s", exc)

    async def stop(self) -> None:
        if getattr(self, "_janitor", None) is not None:
            self._janitor.cancel()
            self._janitor = None
        # drain async writes, then persist everything synchronously
        if self.queue is not None:
            await self.queue.close()
        for part in (self.gpu, self.cpu, self.ports):
            if part is not None:
                await part.persist()
        for part in (
            self.container_versions,
            self.volume_versions,
            self.merges,
            getattr(self, "released", None),
        ):
            if part is not None:
                await part.persist()
        if self.runtime is not None:
            await self.runtime.close()
        if self.store is not None:
            await self.store.close()


def build_app(cfg: Optional[Config] = None, daemon: Optional[Daemon] = None) -> FastAPI:
    cfg = cfg or Config()
    d = daemon or Daemon(cfg)

    @asynccontextmanager
    async def lifespan(app: FastAPI):
        if d.replicaset is None:
            await d.start()
        app.state.daemon = d
        _mount(app, d)
        yield
        await d.stop()

    app = FastAPI(title="gpu-docker-api-amd", lifespan=lifespan)
    app.add_middleware(CorsMiddleware)
    app.add_middleware(AuthMiddleware, apikey=cfg.apikey)

    @app.get("/ping")
    async def ping():
        return {"message": "pong"}

    @app.get("/metrics")
    async def metrics():
        return PlainTextResponse(METRICS.prometheus_text())

    return app


def _mount(app: FastAPI, d: Daemon) -> None:
    app.include_router(replicaset_router.make_router(d.replicaset))
    app.include_router(volume_router.make_router(d.volume))
    app.include_router(resource_router.make_router(d.gpu, d.cpu, d.ports))

    from .response import error, success
    from .codes import Code

    @app.get("/api/v1/events")
    async def events():
        """MI355X extension: live state-change stream (SSE). The reference's
        clients must poll; this streams every container/volume/scheduler
        state mutation as it commits (memory backend only)."""
        store = d.store
        if not hasattr(store, "subscribe"):
            from .codes import Code as _Code
            from .response import error as _error

            return _error(_Code.SERVER_BUSY, detail="state backend does not stream events")

        import asyncio
        import json as _json

        from fastapi.responses import StreamingResponse

        q, unsubscribe = store.subscribe()

        async def gen():
            try:
                yield ": connected\n\n"
                while True:
                    try:
                        ev = await asyncio.wait_for(q.get(), timeout=15.0)
                        yield f"data: {_json.dumps(ev)}\n\n"
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
            finally:
                unsubscribe()

        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.get("/api/v1/images")
    async def images_list():
        """MI355X extension: the runtime's image store (proc/mock: seed
        dirs + commit snapshots; docker: the engine's image list)."""
        try:
            data = await d.runtime.image_list()
        except Exception as exc:  # noqa: BLE001
            return error(Code.SERVER_BUSY, detail=str(exc))
        return success(data)

    @app.put("/api/v1/images/{ref}")
    async def images_import(ref: str, request: Request):
        """MI355X extension: register a local directory as an image — the
        'pull it locally first' step the reference delegates to dockerd
        (its OpenAPI: imageName 'is not automatically downloaded')."""
        import re as _re

        # image refs: name[:tag] with registry-ish chars; traversal-safe —
        # the runtime's _image_dir additionally realpath-contains the
        # flattened ref
        if not _re.fullmatch(r"[A-Za-z0-9_][A-Za-z0-9_.:/\-]{0,127}", ref) or ".." in ref:
            return error(Code.INVALID_PARAMS, detail=f"invalid image ref {ref!r}")
        try:
            body = await request.json()
        except Exception:
            body = {}
        if not isinstance(body, dict):  # fuzz-found: a bare JSON scalar crashed
            return error(Code.INVALID_PARAMS, detail="body must be a JSON object")
        path = body.get("path", "")
        if not path:
            return error(Code.INVALID_PARAMS, detail="body needs {path: <local dir>}")
        try:
            out = await d.runtime.image_import(ref, path)
        except FileNotFoundError:
            return error(Code.INVALID_PARAMS, detail=f"not a directory: {path}")
        except Exception as exc:  # noqa: BLE001
            return error(Code.SERVER_BUSY, detail=str(exc))
        return success({"ref": out})

    @app.post("/api/v1/admin/compact")
    async def compact(request: Request):
        """MI355X extension: discard state history below a revision and
        shrink the WAL (memory backend only). Destroys rollback targets
        below the point — an explicit operator action."""
        try:
            body = await request.json()
        except Exception:
            body = {}
        if not isinstance(body, dict):
            return error(Code.INVALID_PARAMS)
        store = d.store
        if not hasattr(store, "compact"):
            return error(Code.SERVER_BUSY, detail="state backend does not support compaction")
        try:
            rev = int(body.get("revision", 0) or 0)
        except (TypeError, ValueError):
            return error(Code.INVALID_PARAMS)
        try:
            result = await store.compact(rev)
        except Exception as exc:  # noqa: BLE001
            return error(Code.SERVER_BUSY, detail=str(exc))
        return success(result)
