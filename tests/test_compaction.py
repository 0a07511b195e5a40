"""State compaction: WAL shrinks, live state and post-compaction versioning
survive daemon restarts, history below the point is refused."""
import os

import pytest

from gpu_docker_api_amd.state import MemoryStore, Resource
from gpu_docker_api_amd.xerrors import NotExistInStore, RevisionCompacted


def test_compact_shrinks_wal_and_preserves_state(tmp_path, run):
    async def main():
        p = str(tmp_path / "wal")
        s = MemoryStore(wal_path=p)
        for i in range(50):
            await s.put(Resource.CONTAINERS, "app", f"v{i}")
        size_before = os.path.getsize(p)
        out = await s.compact()
        assert out["wal_bytes"] < size_before / 5
        # live state preserved, version numbering intact
        kv = await s.get(Resource.CONTAINERS, "app")
        assert kv.value == "v49"
        assert kv.version == 50
        hist = await s.history(Resource.CONTAINERS, "app")
        assert [h.value for h in hist] == ["v49"]
        await s.close()

        # restart: version numbering must survive the rewritten WAL
        s2 = MemoryStore(wal_path=p)
        kv = await s2.get(Resource.CONTAINERS, "app")
        assert kv.version == 50 and kv.value == "v49"
        await s2.put(Resource.CONTAINERS, "app", "v50")
        assert (await s2.get(Resource.CONTAINERS, "app")).version == 51
        assert (await s2.get_version(Resource.CONTAINERS, "app", 50)).value == "v49"
        await s2.close()

    run(main())


def test_compact_via_admin_endpoint(tmp_path, run):
    from fastapi.testclient import TestClient

    from gpu_docker_api_amd.routers.app import build_app
    from helpers import make_config

    app = build_app(make_config(tmp_path))
    with TestClient(app) as c:
        body = {
            "imageName": "img",
            "replicaSetName": "web",
            "gpuCount": 1,
        }
        c.post("/api/v1/replicaSet", json=body)
        for _ in range(3):
            c.patch("/api/v1/replicaSet/web", json={"memoryPatch": {"memory": "2GB"}})
            c.patch("/api/v1/replicaSet/web", json={"memoryPatch": {"memory": "1GB"}})
        r = c.post("/api/v1/admin/compact", json={})
        assert r.json()["code"] == 200
        assert r.json()["data"]["compacted_revision"] > 0
        # newest state still served; deep history gone
        hist = c.get("/api/v1/replicaSet/web/history").json()["data"]
        assert len(hist) == 1
        info = c.get("/api/v1/replicaSet/web").json()
        assert info["code"] == 200
