"""CORS + static-bearer auth middleware (reference: routers/cors.go:9-32,
routers/auth.go:9-26 — auth disabled when APIKEY env is unset).

Implemented as raw ASGI wrappers, not starlette BaseHTTPMiddleware: the
latter re-buffers every request/response through an ASGI sub-app and costs
~1-2 ms per request — measurable against this daemon's ~1-4 ms operations.
"""
from __future__ import annotations

import hmac

import json

_CORS_HEADERS = [
    (b"access-control-allow-credentials", b"true"),
    (
        b"access-control-allow-headers",
        b"Content-Type, Content-Length, Authorization, Origin, X-Requested-With",
    ),
    (b"access-control-allow-methods", b"GET, POST, PUT, PATCH, DELETE, OPTIONS"),
]


class CorsMiddleware:
    def __init__(self, app) -> None:
        self.app = app

    async def __call__(self, scope, receive, send):
        if scope["type"] != "http":
            return await self.app(scope, receive, send)
        origin = b"*"
        for k, v in scope.get("headers", []):
            if k == b"origin":
                origin = v
                break
        extra = [(b"access-control-allow-origin", origin)] + _CORS_HEADERS

        if scope.get("method") == "OPTIONS":
            await send(
                {"type": "http.response.start", "status": 204, "headers": extra}
            )
            await send({"type": "http.response.body", "body": b""})
            return

        async def send_with_cors(message):
            if message["type"] == "http.response.start":
                message = dict(message)
                message["headers"] = list(message.get("headers", [])) + extra
            await send(message)

        await self.app(scope, receive, send_with_cors)


class AuthMiddleware:
    def __init__(self, app, apikey: str = "") -> None:
        self.app = app
        self.apikey = apikey.encode()

    async def __call__(self, scope, receive, send):
        if scope["type"] != "http" or not self.apikey:
            return await self.app(scope, receive, send)
        path = scope.get("path", "")
        if path in ("/ping", "/metrics"):
            return await self.app(scope, receive, send)
        token = b""
        for k, v in scope.get("headers", []):
            if k == b"authorization":
                token = v[7:] if v[:7].lower() == b"bearer " else v
                break
        # constant-time compare: a == on secrets leaks length/prefix timing
        if not hmac.compare_digest(token, self.apikey):
            body = json.dumps(
                {"code": 403, "msg": "Forbidden", "data": None}
            ).encode()
            await send(
                {
                    "type": "http.response.start",
                    "status": 200,  # envelope always HTTP 200 (reference contract)
                    "headers": [(b"content-type", b"application/json")],
                }
            )
            await send({"type": "http.response.body", "body": body})
            return
        await self.app(scope, receive, send)
