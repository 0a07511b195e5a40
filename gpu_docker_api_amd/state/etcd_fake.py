"""In-process etcd v3 JSON-gateway server backed by MemoryMVCC.

Serves the same three endpoints EtcdGatewayStore speaks (/v3/kv/put, range,
deleterange) with etcd's wire conventions (base64 keys/values, string
integers, MVCC revision reads). Used (a) as the MVCC-faithful fake for
gateway-client tests — SURVEY.md §4 notes the revision walker *requires* an
MVCC-faithful fake — and (b) as a dev-mode stand-in for a real etcd.
"""
from __future__ import annotations

import base64

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from ..xerrors import RevisionCompacted
from .mvcc import MemoryMVCC


def _b64(s: str) -> str:
    return base64.b64encode(s.encode()).decode()


def _unb64(s: str) -> str:
    return base64.b64decode(s).decode()


def build_fake_etcd(store: MemoryMVCC | None = None) -> FastAPI:
    mvcc = store if store is not None else MemoryMVCC()
    app = FastAPI(title="fake-etcd-gateway")
    app.state.mvcc = mvcc

    def header() -> dict:
        return {"revision": str(mvcc.revision)}

    def kv_json(kv) -> dict:
        return {
            "key": _b64(kv.key),
            "value": _b64(kv.value),
            "create_revision": str(kv.create_revision),
            "mod_revision": str(kv.mod_revision),
            "version": str(kv.version),
        }

    @app.post("/v3/kv/put")
    async def put(request: Request):
        body = await request.json()
        mvcc.put(_unb64(body["key"]), _unb64(body.get("value", "")))
        return {"header": header()}

    @app.post("/v3/kv/range")
    async def range_(request: Request):
        body = await request.json()
        key = _unb64(body["key"])
        rev = int(body.get("revision", 0) or 0)
        try:
            if "range_end" in body and body["range_end"]:
                # prefix ranges only (what the client uses)
                kvs = mvcc.range_prefix(key, rev=rev)
            else:
                kv = mvcc.get_or_none(key) if rev == 0 else None
                if rev:
                    try:
                        kv = mvcc.get(key, rev=rev)
                    except Exception as exc:
                        if isinstance(exc, RevisionCompacted):
                            raise
                        kv = None
                kvs = [kv] if kv is not None else []
        except RevisionCompacted:
            return JSONResponse(
                status_code=400,
                content={
                    "code": 11,
                    "message": "etcdserver: mvcc: required revision has been compacted",
                },
            )
        return {
            "header": header(),
            "kvs": [kv_json(kv) for kv in kvs],
            "count": str(len(kvs)),
        }

    @app.post("/v3/kv/compaction")
    async def compaction(request: Request):
        body = await request.json()
        rev = int(body.get("revision", 0) or 0)
        try:
            mvcc.compact(rev)
        except RevisionCompacted:
            return JSONResponse(
                status_code=400,
                content={"code": 11,
                         "message": "etcdserver: mvcc: required revision has been compacted"},
            )
        return {"header": header()}

    @app.post("/v3/kv/deleterange")
    async def deleterange(request: Request):
        body = await request.json()
        key = _unb64(body["key"])
        if "range_end" in body and body["range_end"]:
            deleted = mvcc.delete_prefix(key)
        else:
            deleted = mvcc.delete(key)
        return {"header": header(), "deleted": str(deleted)}

    return app


def main() -> None:
    """Serve the gateway on TCP for out-of-process integration tests / dev:
    ``python -m gpu_docker_api_amd.state.etcd_fake --port 2379``."""
    import argparse

    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, required=True)
    args = p.parse_args()
    uvicorn.run(build_fake_etcd(), host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
