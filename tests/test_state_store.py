"""StateStore facade + WorkQueue write-behind semantics."""
import asyncio

import pytest

from gpu_docker_api_amd.state import (
    DelKey,
    MemoryStore,
    PutKeyValue,
    Resource,
    WorkQueue,
)
from gpu_docker_api_amd.xerrors import NotExistInStore


def test_memory_store_roundtrip(run):
    async def main():
        s = MemoryStore()
        await s.put(Resource.CONTAINERS, "foo", "v1")
        await s.put(Resource.CONTAINERS, "foo", "v2")
        kv = await s.get(Resource.CONTAINERS, "foo")
        assert kv.value == "v2"
        assert kv.key == "/gpu-docker-api/apis/v1/containers/foo"
        hist = await s.history(Resource.CONTAINERS, "foo")
        assert [h.value for h in hist] == ["v2", "v1"]
        assert (await s.get_version(Resource.CONTAINERS, "foo", 1)).value == "v1"
        assert await s.get_or_none(Resource.VOLUMES, "foo") is None
        assert await s.delete(Resource.CONTAINERS, "foo") == 1
        with pytest.raises(NotExistInStore):
            await s.get(Resource.CONTAINERS, "foo")

    run(main())


def test_memory_store_wal_persistence(tmp_path, run):
    async def main():
        p = str(tmp_path / "state.wal")
        s = MemoryStore(wal_path=p)
        await s.put(Resource.VOLUMES, "vol", "spec1")
        await s.put(Resource.VOLUMES, "vol", "spec2")
        await s.close()
        s2 = MemoryStore(wal_path=p)
        assert (await s2.get(Resource.VOLUMES, "vol")).value == "spec2"
        assert len(await s2.history(Resource.VOLUMES, "vol")) == 2

    run(main())


def test_workqueue_flushes_and_drains(run):
    async def main():
        s = MemoryStore()
        q = WorkQueue(s)
        q.start()
        q.put(Resource.GPUS, "gpuStatusMapKey", "{}")
        q.delete(Resource.GPUS, "missing")  # delete of absent key is fine
        await q.drain()
        assert (await s.get(Resource.GPUS, "gpuStatusMapKey")).value == "{}"
        await q.close()

    run(main())


def test_workqueue_retries_transient_failures(run):
    class Flaky(MemoryStore):
        def __init__(self):
            super().__init__()
            self.fails = 2

        async def put(self, resource, key, value):
            if self.fails > 0:
                self.fails -= 1
                raise RuntimeError("transient")
            await super().put(resource, key, value)

    async def main():
        s = Flaky()
        q = WorkQueue(s, max_attempts=5)
        q.start()
        q.put(Resource.CPUS, "cpuStatusMapKey", "{}")
        for _ in range(200):
            await asyncio.sleep(0.01)
            if await s.get_or_none(Resource.CPUS, "cpuStatusMapKey"):
                break
        assert (await s.get(Resource.CPUS, "cpuStatusMapKey")).value == "{}"
        await q.close()

    run(main())


def test_workqueue_drops_poisoned_after_max_attempts(run):
    class Broken(MemoryStore):
        async def put(self, resource, key, value):
            raise RuntimeError("permanent")

    async def main():
        q = WorkQueue(Broken(), max_attempts=2)
        q.start()
        q.put(Resource.PORTS, "usedPortSetKey", "{}")
        await asyncio.sleep(0.3)  # retry backoff elapses; item is dropped
        await q.close()           # close() must not hang on the poisoned item

    run(main())
