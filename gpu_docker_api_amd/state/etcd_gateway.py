"""etcd v3 JSON-gateway backend for StateStore.

Speaks the grpc-gateway HTTP/JSON endpoints every etcd >= 3.4 serves on its
client port (``/v3/kv/put``, ``/v3/kv/range``, ``/v3/kv/deleterange``), so no
generated gRPC stubs are needed. Key scheme and value shapes are identical to
the reference's clientv3 usage (/root/reference/internal/etcd/common.go:15-68),
making this a drop-in against an existing deployment.

History here must walk real etcd MVCC revisions exactly as the reference does
(/root/reference/internal/etcd/revision.go:18-66): one ranged read per
revision from mod_revision down to create_revision, deduped by the per-key
``version`` counter. (The MemoryStore backend answers the same query in one
call — this walker is the compatibility path.)
"""
from __future__ import annotations

import base64
from typing import Any, Dict, List, Optional

import httpx

from ..xerrors import NotExistInStore, RevisionCompacted
from .keys import Resource, resource_key, resource_prefix
from .mvcc import KeyValue
from .store import StateStore


def _b64(s: str) -> str:
    return base64.b64encode(s.encode()).decode()


def _unb64(s: str) -> str:
    return base64.b64decode(s).decode()


def _prefix_range_end(prefix: str) -> str:
    b = bytearray(prefix.encode())
    for i in reversed(range(len(b))):
        if b[i] < 0xFF:
            b[i] += 1
            del b[i + 1 :]
            return base64.b64encode(bytes(b)).decode()
    return base64.b64encode(b"\x00").decode()  # whole keyspace


def _kv_from_json(kv: Dict[str, Any]) -> KeyValue:
    return KeyValue(
        key=_unb64(kv["key"]),
        value=_unb64(kv.get("value", "")),
        create_revision=int(kv.get("create_revision", 0)),
        mod_revision=int(kv.get("mod_revision", 0)),
        version=int(kv.get("version", 0)),
    )


class EtcdGatewayStore(StateStore):
    def __init__(
        self,
        endpoint: str = "http://127.0.0.1:2379",
        timeout: float = 2.0,
        transport: Optional[httpx.AsyncBaseTransport] = None,
    ) -> None:
        # reference: dial timeout 2s (etcd/client.go:17), op timeout 1s (common.go:31)
        self.endpoint = endpoint.rstrip("/")
        self._client = httpx.AsyncClient(
            base_url=self.endpoint, timeout=timeout, transport=transport
        )

    async def _call(self, path: str, body: Dict[str, Any]) -> Dict[str, Any]:
        resp = await self._client.post(path, json=body)
        if resp.status_code != 200:
            try:
                detail = resp.json()
            except Exception:
                detail = {"message": resp.text}
            msg = str(detail.get("message", detail))
            if "compacted" in msg:
                raise RevisionCompacted(msg)
            raise RuntimeError(f"etcd gateway {path}: {resp.status_code} {msg}")
        return resp.json()

    async def put(self, resource: Resource, key: str, value: str) -> None:
        await self._call(
            "/v3/kv/put", {"key": _b64(resource_key(resource, key)), "value": _b64(value)}
        )

    async def _range(
        self, key: str, range_end: Optional[str] = None, revision: int = 0
    ) -> List[KeyValue]:
        body: Dict[str, Any] = {"key": _b64(key)}
        if range_end is not None:
            body["range_end"] = range_end
        if revision:
            body["revision"] = revision
        data = await self._call("/v3/kv/range", body)
        return [_kv_from_json(kv) for kv in data.get("kvs", [])]

    async def get(self, resource: Resource, key: str) -> KeyValue:
        kvs = await self._range(resource_key(resource, key))
        if not kvs:
            raise NotExistInStore(resource_key(resource, key))
        return kvs[0]

    async def delete(self, resource: Resource, key: str) -> int:
        data = await self._call(
            "/v3/kv/deleterange", {"key": _b64(resource_key(resource, key))}
        )
        return int(data.get("deleted", 0))

    async def range(self, resource: Resource) -> List[KeyValue]:
        prefix = resource_prefix(resource) + "/"
        return await self._range(prefix, range_end=_prefix_range_end(prefix))

    # Safety valve: the walk below is O(#writes of the key); this bounds a
    # pathological key (e.g. shared with a high-churn external writer).
    MAX_WALK_STEPS = 4096

    async def _walk(self, full_key: str):
        """Yield each put of the key's current lifetime, newest first.

        The reference decrements the STORE-WIDE revision by one per round
        trip (revision.go:18-44) — O(total revisions). etcd's answer to
        ``range(key, revision=r)`` carries the kv's actual mod_revision
        (the last put at or below r), so stepping to ``mod_revision - 1``
        jumps straight to the previous put: O(#writes of this key) round
        trips, identical result set."""
        head = await self._range(full_key)
        if not head:
            raise NotExistInStore(full_key)
        cur = head[0]
        rev = cur.mod_revision
        steps = 0
        while rev >= cur.create_revision and steps < self.MAX_WALK_STEPS:
            steps += 1
            try:
                kvs = await self._range(full_key, revision=rev)
            except RevisionCompacted:
                return
            if not kvs:
                return
            yield kvs[0]
            rev = kvs[0].mod_revision - 1

    async def history(self, resource: Resource, key: str) -> List[KeyValue]:
        """The reference's GetRevisionRange result (revision.go:18-44)."""
        out: List[KeyValue] = []
        seen_versions = set()
        async for kv in self._walk(resource_key(resource, key)):
            if kv.version not in seen_versions:
                seen_versions.add(kv.version)
                out.append(kv)
        return out

    async def get_version(self, resource: Resource, key: str, version: int) -> KeyValue:
        """The reference's GetRevision(version) result (revision.go:46-66)."""
        full_key = resource_key(resource, key)
        async for kv in self._walk(full_key):
            if kv.version == version:
                return kv
        raise NotExistInStore(f"{full_key} version={version}")

    async def close(self) -> None:
        await self._client.aclose()
