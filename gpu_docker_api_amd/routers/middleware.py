"""CORS + static-bearer auth middleware (reference: routers/cors.go:9-32,
routers/auth.go:9-26 — auth disabled when APIKEY env is unset)."""
from __future__ import annotations

from starlette.middleware.base import BaseHTTPMiddleware
from starlette.requests import Request

from .codes import Code
from .response import error


class CorsMiddleware(BaseHTTPMiddleware):
    async def dispatch(self, request: Request, call_next):
        origin = request.headers.get("origin", "*")
        if request.method == "OPTIONS":
            from starlette.responses import Response

            resp = Response(status_code=204)
        else:
            resp = await call_next(request)
        resp.headers["Access-Control-Allow-Origin"] = origin
        resp.headers["Access-Control-Allow-Credentials"] = "true"
        resp.headers["Access-Control-Allow-Headers"] = (
            "Content-Type, Content-Length, Authorization, Origin, X-Requested-With"
        )
        resp.headers["Access-Control-Allow-Methods"] = (
            "GET, POST, PUT, PATCH, DELETE, OPTIONS"
        )
        return resp


class AuthMiddleware(BaseHTTPMiddleware):
    def __init__(self, app, apikey: str = "") -> None:
        super().__init__(app)
        self.apikey = apikey

    async def dispatch(self, request: Request, call_next):
        if not self.apikey or request.url.path in ("/ping", "/metrics"):
            return await call_next(request)
        auth = request.headers.get("authorization", "")
        token = auth[7:] if auth.lower().startswith("bearer ") else auth
        if token != self.apikey:
            return error(Code.FORBIDDEN)
        return await call_next(request)
