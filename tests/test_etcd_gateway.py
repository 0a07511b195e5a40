"""EtcdGatewayStore against the MVCC-faithful fake gateway: the reference's
revision-walker semantics must hold over the wire (etcd/revision.go:18-66)."""
import httpx
import pytest

from gpu_docker_api_amd.state.etcd_fake import build_fake_etcd
from gpu_docker_api_amd.state.etcd_gateway import EtcdGatewayStore
from gpu_docker_api_amd.state.keys import Resource
from gpu_docker_api_amd.xerrors import NotExistInStore


def make_store():
    app = build_fake_etcd()
    transport = httpx.ASGITransport(app=app)
    return EtcdGatewayStore("http://fake-etcd", transport=transport), app


def test_put_get_delete_over_gateway(run):
    async def main():
        store, _ = make_store()
        await store.put(Resource.CONTAINERS, "foo", "v1")
        kv = await store.get(Resource.CONTAINERS, "foo")
        assert kv.value == "v1"
        assert kv.key == "/gpu-docker-api/apis/v1/containers/foo"
        assert kv.version == 1
        assert await store.delete(Resource.CONTAINERS, "foo") == 1
        with pytest.raises(NotExistInStore):
            await store.get(Resource.CONTAINERS, "foo")
        await store.close()

    run(main())


def test_history_walks_revisions(run):
    async def main():
        store, app = make_store()
        await store.put(Resource.CONTAINERS, "foo", "v1")
        await store.put(Resource.VOLUMES, "other", "x")  # interleaved writes
        await store.put(Resource.CONTAINERS, "foo", "v2")
        await store.put(Resource.VOLUMES, "other", "y")
        await store.put(Resource.CONTAINERS, "foo", "v3")
        hist = await store.history(Resource.CONTAINERS, "foo")
        assert [h.value for h in hist] == ["v3", "v2", "v1"]
        assert [h.version for h in hist] == [3, 2, 1]
        kv = await store.get_version(Resource.CONTAINERS, "foo", 2)
        assert kv.value == "v2"
        with pytest.raises(NotExistInStore):
            await store.get_version(Resource.CONTAINERS, "foo", 9)
        await store.close()

    run(main())


def test_history_stops_at_compaction(run):
    async def main():
        store, app = make_store()
        mvcc = app.state.mvcc
        await store.put(Resource.CONTAINERS, "foo", "v1")
        await store.put(Resource.CONTAINERS, "foo", "v2")
        await store.put(Resource.CONTAINERS, "foo", "v3")
        mvcc.compact(mvcc.revision)  # only the newest state survives
        hist = await store.history(Resource.CONTAINERS, "foo")
        assert [h.value for h in hist] == ["v3"]
        await store.close()

    run(main())


def test_range_prefix(run):
    async def main():
        store, _ = make_store()
        await store.put(Resource.CONTAINERS, "b", "2")
        await store.put(Resource.CONTAINERS, "a", "1")
        await store.put(Resource.VOLUMES, "v", "3")
        kvs = await store.range(Resource.CONTAINERS)
        assert [kv.value for kv in kvs] == ["1", "2"]
        await store.close()

    run(main())


def test_full_daemon_on_etcd_backend(tmp_path, run):
    """The whole control plane runs against the etcd gateway backend."""

    async def main():
        from gpu_docker_api_amd.models import ContainerRun, GpuPatch, PatchRequest
        from gpu_docker_api_amd.routers.app import Daemon
        from helpers import make_config

        app = build_fake_etcd()
        transport = httpx.ASGITransport(app=app)
        store = EtcdGatewayStore("http://fake-etcd", transport=transport)
        d2 = Daemon(make_config(tmp_path), store=store)
        await d2.start()
        out = await d2.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="etcdtest", gpu_count=1)
        )
        assert out["name"] == "etcdtest-1"
        await d2.replicaset.patch_container(
            "etcdtest", PatchRequest(gpu_patch=GpuPatch(gpu_count=2))
        )
        hist = await d2.replicaset.get_container_history("etcdtest")
        assert [h["version"] for h in hist] == [2, 1]
        # the fake's MVCC holds the reference key scheme
        kvs = app.state.mvcc.range_prefix("/gpu-docker-api/apis/v1/containers/")
        assert len(kvs) == 1
        await d2.stop()

    run(main())
