"""Volume routes — 5 endpoints under /api/v1/volumes
(reference: internal/routers/volume.go:19-25; validation :37-47, :84-90)."""
from __future__ import annotations

from fastapi import APIRouter, Request
from pydantic import ValidationError

from ..models import VolumeCreate, VolumeSize
from ..models.memory import parse_size
from ..utils.names import valid_name
from ..services.volume import VolumeService
from .codes import Code
from .errors import log_error, map_error
from .response import error, success


def make_router(svc: VolumeService) -> APIRouter:
    r = APIRouter(prefix="/api/v1/volumes")

    async def _parse(request: Request, model):
        try:
            return model.model_validate(await request.json())
        except (ValidationError, ValueError):
            return None

    def _valid_size(s: str) -> bool:
        try:
            parse_size(s)
            return True
        except Exception:
            return False

    def _bad_name(name: str):
        """Strict identifier grammar: the reference's no-dash/no-leading-slash
        check (volume.go:37-47) still admits '..' and 'a/b', which traverse
        out of the volumes dir on the proc runtime (utils/names.py)."""
        if not name:
            return error(Code.VOLUME_NAME_EMPTY)
        if "-" in name:
            return error(Code.VOLUME_NAME_DASH)
        if name.startswith("/"):
            return error(Code.VOLUME_NAME_SLASH)
        if not valid_name(name):
            return error(Code.INVALID_PARAMS, detail=f"invalid volume name: {name!r}")
        return None

    @r.post("")
    async def create(request: Request):
        req = await _parse(request, VolumeCreate)
        if req is None:
            return error(Code.INVALID_PARAMS)
        bad = _bad_name(req.name)
        if bad is not None:
            return bad
        if req.size and not _valid_size(req.size):
            return error(Code.VOLUME_SIZE_UNIT)
        try:
            data = await svc.create_volume(req)
        except Exception as exc:  # noqa: BLE001
            log_error("volume.create", exc)
            return error(map_error(exc, Code.VOLUME_CREATE_FAILED, volume=True))
        return success(data)

    @r.get("")
    async def list_all():
        """MI355X extension: list all volumes."""
        try:
            data = await svc.list_volumes()
        except Exception as exc:  # noqa: BLE001
            log_error("volume.list", exc)
            return error(map_error(exc, Code.VOLUME_GET_INFO_FAILED, volume=True))
        return success(data)

    @r.patch("/{name}/size")
    async def patch_size(name: str, request: Request):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        req = await _parse(request, VolumeSize)
        if req is None:
            return error(Code.INVALID_PARAMS)
        if not req.size or not _valid_size(req.size):
            return error(Code.VOLUME_SIZE_UNIT)
        try:
            data = await svc.patch_volume_size(name, req.size)
        except Exception as exc:  # noqa: BLE001
            log_error("volume.patch", exc)
            return error(map_error(exc, Code.VOLUME_PATCH_FAILED, volume=True))
        return success(data)

    @r.delete("/{name}")
    async def delete(name: str, request: Request):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        # ?noall present => keep the store record (reference volume.go:123-129)
        keep = "noall" in request.query_params
        try:
            await svc.delete_volume(name, keep_record=keep)
        except Exception as exc:  # noqa: BLE001
            log_error("volume.delete", exc)
            return error(map_error(exc, Code.VOLUME_DELETE_FAILED, volume=True))
        return success(None)

    @r.get("/{name}")
    async def info(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            data = await svc.get_volume_info(name)
        except Exception as exc:  # noqa: BLE001
            log_error("volume.info", exc)
            return error(map_error(exc, Code.VOLUME_GET_INFO_FAILED, volume=True))
        return success(data)

    @r.get("/{name}/history")
    async def history(name: str):
        bad = _bad_name(name)
        if bad is not None:
            return bad
        try:
            data = await svc.get_volume_history(name)
        except Exception as exc:  # noqa: BLE001
            log_error("volume.history", exc)
            return error(map_error(exc, Code.VOLUME_GET_HISTORY_FAILED, volume=True))
        return success(data)

    return r
