"""Uniform response envelope: always HTTP 200, ``{code, msg, data}``
(reference: internal/routers/response.go:9-29)."""
from __future__ import annotations

from typing import Any, Optional

from fastapi.responses import JSONResponse

from .codes import Code, msg


def success(data: Any = None) -> JSONResponse:
    return JSONResponse(
        status_code=200,
        content={"code": int(Code.SUCCESS), "msg": msg(Code.SUCCESS), "data": data},
    )


def error(code: Code, data: Any = None, detail: Optional[str] = None) -> JSONResponse:
    body = {"code": int(code), "msg": msg(code), "data": data}
    if detail:
        body["detail"] = detail
    return JSONResponse(status_code=200, content=body)
