"""ReplicaSet routes — 12 endpoints under /api/v1/replicaSet
(reference: internal/routers/replicaset.go:22-57; validation :110-148)."""
from __future__ import annotations

from fastapi import APIRouter, Request
from pydantic import ValidationError

from ..models import (
    ContainerCommit,
    ContainerExecute,
    ContainerRun,
    PatchRequest,
    RollbackRequest,
)
from ..models.memory import parse_size
from ..services.replicaset import ReplicaSetService
from ..utils.names import valid_name
from .codes import Code
from .errors import log_error, map_error
from .response import error, success


def _valid_memory(s: str) -> bool:
    try:
        parse_size(s)
        return True
    except Exception:
        return False


def _bad_name(name: str):
    """Reject names that are not strict identifiers before they reach any
    filesystem join (a name like '..' or 'a/b' in the reference's weaker
    validation would traverse out of the data dir — see utils/names.py)."""
    if not name:
        return error(Code.CONTAINER_NAME_EMPTY)
    if "-" in name:
        return error(Code.CONTAINER_NAME_DASH)
    if not valid_name(name):
        return error(Code.INVALID_PARAMS, detail=f"invalid replicaSet name: {name!r}")
    return None


def make_router(svc: ReplicaSetService) -> APIRouter:
    r = APIRouter(prefix="/api/v1/replicaSet")

    async def _parse(request: Request, model):
        try:
            return model.model_validate(await request.json())
        except (ValidationError, ValueError):
            return None

    @r.post("")
    async def run(request: Request):
        req = await _parse(request, ContainerRun)
        if req is None:
            return error(Code.INVALID_PARAMS)
        if not req.image_name:
            return error(Code.IMAGE_NAME_EMPTY)
        bad = _bad_name(req.replica_set_name)
        if bad is not None:
            return bad
        if req.gpu_count < 0:
            return error(Code.GPU_COUNT_GE_ZERO)
        if req.cpu_count < 0:
            return error(Code.CPU_COUNT_GE_ZERO)
        if req.memory and not _valid_memory(req.memory):
            return error(Code.CONTAINER_MEMORY_UNIT)
        try:
            data = await svc.run_gpu_container(req)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.run", exc)
            return error(map_error(exc, Code.CONTAINER_RUN_FAILED))
        return success(data)

    @r.get("")
    async def list_all():
        """MI355X extension: list all replicaSets (the reference has no
        list route; its clients must track names externally)."""
        try:
            data = await svc.list_replicasets()
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.list", exc)
            return error(map_error(exc, Code.CONTAINER_GET_INFO_FAILED))
        return success(data)

    @r.get("/{name}")
    async def info(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            data = await svc.get_container_info(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.info", exc)
            return error(map_error(exc, Code.CONTAINER_GET_INFO_FAILED))
        return success(data)

    @r.get("/{name}/logs")
    async def logs(name: str, request: Request):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            tail = min(10000, max(1, int(request.query_params.get("tail", "200"))))
        except ValueError:
            return error(Code.INVALID_PARAMS)
        try:
            data = await svc.get_container_logs(name, tail=tail)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.logs", exc)
            return error(map_error(exc, Code.CONTAINER_GET_INFO_FAILED))
        return success({"logs": data})

    @r.get("/{name}/stats")
    async def stats(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            data = await svc.get_container_stats(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.stats", exc)
            return error(map_error(exc, Code.CONTAINER_GET_INFO_FAILED))
        return success(data)

    @r.get("/{name}/history")
    async def history(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            data = await svc.get_container_history(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.history", exc)
            return error(map_error(exc, Code.CONTAINER_GET_HISTORY_FAILED))
        return success(data)

    @r.post("/{name}/commit")
    async def commit(name: str, request: Request):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        req = await _parse(request, ContainerCommit)
        if req is None:
            return error(Code.INVALID_PARAMS)
        if not req.new_image_name:
            # the reference would tag an empty string (replicaset.go:883-890)
            return error(Code.IMAGE_NAME_EMPTY)
        try:
            image = await svc.commit_container(name, req)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.commit", exc)
            return error(map_error(exc, Code.CONTAINER_COMMIT_FAILED))
        return success({"imageName": image})

    @r.post("/{name}/execute")
    async def execute(name: str, request: Request):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        req = await _parse(request, ContainerExecute)
        if req is None:
            return error(Code.INVALID_PARAMS)
        if not req.cmd:
            return error(Code.INVALID_PARAMS)
        try:
            stdout, rc = await svc.execute_container(name, req)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.execute", exc)
            return error(map_error(exc, Code.CONTAINER_EXECUTE_FAILED))
        return success({"stdout": stdout, "exitCode": rc})

    @r.patch("/{name}")
    async def patch(name: str, request: Request):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        req = await _parse(request, PatchRequest)
        if req is None:
            return error(Code.INVALID_PARAMS)
        if req.gpu_patch is not None and req.gpu_patch.gpu_count < 0:
            return error(Code.GPU_COUNT_GE_ZERO)
        if req.cpu_patch is not None and req.cpu_patch.cpu_count < 0:
            return error(Code.CPU_COUNT_GE_ZERO)
        if (
            req.memory_patch is not None
            and req.memory_patch.memory
            and not _valid_memory(req.memory_patch.memory)
        ):
            return error(Code.CONTAINER_MEMORY_UNIT)
        try:
            data = await svc.patch_container(name, req)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.patch", exc)
            return error(map_error(exc, Code.CONTAINER_PATCH_FAILED))
        return success(data)

    @r.patch("/{name}/rollback")
    async def rollback(name: str, request: Request):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        req = await _parse(request, RollbackRequest)
        if req is None:
            return error(Code.INVALID_PARAMS)
        if req.version < 0:
            return error(Code.CONTAINER_VERSION_GE_ZERO)
        try:
            data = await svc.rollback_container(name, req.version, req.restore_data)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.rollback", exc)
            return error(map_error(exc, Code.CONTAINER_ROLLBACK_FAILED))
        return success(data)

    @r.patch("/{name}/stop")
    async def stop(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            await svc.stop_container(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.stop", exc)
            return error(map_error(exc, Code.CONTAINER_STOP_FAILED))
        return success(None)

    @r.patch("/{name}/pause")
    async def pause(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            await svc.pause_container(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.pause", exc)
            return error(map_error(exc, Code.CONTAINER_SHUTDOWN_FAILED))
        return success(None)

    @r.patch("/{name}/continue")
    async def cont(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            await svc.startup_container(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.continue", exc)
            return error(map_error(exc, Code.CONTAINER_STARTUP_FAILED))
        return success(None)

    @r.patch("/{name}/restart")
    async def restart(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            data = await svc.restart_container(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.restart", exc)
            return error(map_error(exc, Code.CONTAINER_RESTART_FAILED))
        return success(data)

    @r.delete("/{name}")
    async def delete(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            await svc.delete_container(name)
        except Exception as exc:  # noqa: BLE001
            log_error("replicaSet.delete", exc)
            return error(map_error(exc, Code.CONTAINER_DELETE_FAILED))
        return success(None)

    return r
