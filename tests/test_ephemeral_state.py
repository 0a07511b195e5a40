"""Ephemeral (no-history) puts: high-churn singleton keys must not grow the
store or the WAL without bound (found by a 5-minute soak: +850 MB RSS)."""
import os

from gpu_docker_api_amd.state import MemoryStore, Resource
from gpu_docker_api_amd.state.mvcc import MemoryMVCC


def test_ephemeral_put_collapses_history():
    s = MemoryMVCC()
    for i in range(1000):
        s.put("/gpus/map", f"state-{i}", retain_history=False)
    # single retained event; version/revision counters still advanced
    assert len(s._hist["/gpus/map"]) == 1
    kv = s.get("/gpus/map")
    assert kv.value == "state-999"
    assert kv.version == 1000
    assert s.revision == 1001
    # normal keys still retain full history
    for i in range(5):
        s.put("/containers/app", f"v{i}")
    assert len(s.history("/containers/app")) == 5


def test_ephemeral_after_delete_recreates_lifetime():
    s = MemoryMVCC()
    s.put("/k", "a", retain_history=False)
    s.delete("/k")
    s.put("/k", "b", retain_history=False)
    kv = s.get("/k")
    assert kv.version == 1  # fresh lifetime after the delete


def test_store_ephemeral_and_wal_rewrite(tmp_path, run):
    async def main():
        p = str(tmp_path / "wal")
        st = MemoryStore(wal_path=p)
        st._wal.max_bytes = 20_000  # force the auto-rewrite path
        for i in range(9000):  # > the 8192-record check interval
            await st.put_ephemeral(Resource.GPUS, "gpuStatusMapKey", f"s{i}")
        size = os.path.getsize(p)
        assert size < 100_000, f"WAL did not shrink: {size} bytes"
        await st.close()
        # restart: latest state survives the rewritten WAL
        st2 = MemoryStore(wal_path=p)
        kv = await st2.get(Resource.GPUS, "gpuStatusMapKey")
        assert kv.value == "s8999"
        await st2.close()

    run(main())


def test_dead_lifetime_pruning_bounds_churn(run):
    async def main():
        s = MemoryStore()
        # 500 create/patch/delete cycles of the same name (soak pattern)
        for i in range(500):
            await s.put(Resource.CONTAINERS, "churn", f"v1-{i}")
            await s.put(Resource.CONTAINERS, "churn", f"v2-{i}")
            await s.delete(Resource.CONTAINERS, "churn")
        key = "/gpu-docker-api/apis/v1/containers/churn"
        # bounded: tombstone + nothing else pending (dead lifetimes pruned)
        assert len(s.mvcc._hist[key]) <= 4  # prev tombstone + last lifetime + tombstone
        # live history still full within a lifetime
        await s.put(Resource.CONTAINERS, "churn", "a")
        await s.put(Resource.CONTAINERS, "churn", "b")
        assert [kv.value for kv in await s.history(Resource.CONTAINERS, "churn")] == ["b", "a"]

    run(main())
