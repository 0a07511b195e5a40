"""GPU/CPU/port schedulers + topology-aware placement + version maps."""
import json

import pytest

from gpu_docker_api_amd.parallel import (
    CpuScheduler,
    GpuScheduler,
    MockInventory,
    PortScheduler,
    Topology,
)
from gpu_docker_api_amd.state import MemoryStore, Resource, WorkQueue
from gpu_docker_api_amd.version import MergeMap, VersionMap
from gpu_docker_api_amd.xerrors import CpuNotEnough, GpuNotEnough, PortNotEnough


def test_gpu_apply_restore_roundtrip(run):
    async def main():
        store = MemoryStore()
        gs = await GpuScheduler.create(store, None, MockInventory(8))
        uuids = gs.apply(3)
        assert len(uuids) == 3
        status = gs.get_gpu_status()
        assert sum(status.values()) == 3
        with pytest.raises(GpuNotEnough):
            gs.apply(6)
        gs.restore(uuids)
        assert sum(gs.get_gpu_status().values()) == 0
        with pytest.raises(GpuNotEnough):
            gs.apply(9)

    run(main())


def test_gpu_state_persists_and_reloads(run):
    async def main():
        store = MemoryStore()
        q = WorkQueue(store)
        q.start()
        gs = await GpuScheduler.create(store, q, MockInventory(4))
        used = gs.apply(2)
        await q.drain()
        kv = await store.get(Resource.GPUS, "gpuStatusMapKey")
        data = json.loads(kv.value)
        assert data["availableGpuNums"] == 4
        assert sum(data["gpuStatusMap"].values()) == 2
        # a fresh scheduler (daemon restart) resumes the allocation state
        gs2 = await GpuScheduler.create(store, q, MockInventory(4))
        assert sum(gs2.get_gpu_status().values()) == 2
        assert all(gs2.get_gpu_status()[u] == 1 for u in used)
        await q.close()

    run(main())


def test_topology_best_subset_prefers_connected():
    # 4 GPUs: 0-1 and 2-3 strongly linked; cross pairs weak.
    mat = [
        [0, 150, 10, 10],
        [150, 0, 10, 10],
        [10, 10, 0, 150],
        [10, 10, 150, 0],
    ]
    t = Topology([[float(x) for x in row] for row in mat], [f"g{i}" for i in range(4)])
    assert t.best_subset([0, 1, 2, 3], 2) in ([0, 1], [2, 3])
    assert t.best_subset([1, 2, 3], 2) == [2, 3]
    # n >= free: take everything
    assert t.best_subset([0, 3], 2) == [0, 3]


def test_topology_overlay_measured():
    t = Topology([[0.0, 50.0], [50.0, 0.0]], ["a", "b"])
    t.overlay_measured({"gpus": ["a", "b"], "p2p_gbps": [[0, 148.5], [147.9, 0]]})
    assert t.measured
    assert t.bandwidth("a", "b") == 148.5


def test_gpu_min_free_hbm_filter(run):
    async def main():
        store = MemoryStore()
        inv = MockInventory(2)
        inv._used[0] = 280 * 1024**3  # GPU 0 nearly full
        gs = await GpuScheduler.create(store, None, inv)
        uuids = gs.apply(1, min_free_hbm=100 * 1024**3)
        assert uuids == ["MockMI355X-1"]

    run(main())


def test_run_with_gpu_memory_floor(tmp_path, run):
    import sys

    sys.path.insert(0, __file__.rsplit("/", 1)[0])
    from helpers import make_daemon

    from gpu_docker_api_amd.models import ContainerRun

    async def main():
        d = await make_daemon(tmp_path)
        # mark GPU 0..6 nearly full; only GPU 7 has 100GB free
        inv = d.gpu.inventory
        for i in range(7):
            inv._used[i] = 250 * 1024**3
        d.gpu.gpus = inv.enumerate()
        out = await d.replicaset.run_gpu_container(
            ContainerRun(
                image_name="img",
                replica_set_name="picky",
                gpu_count=1,
                gpu_memory="100GB",
            )
        )
        st = await d.runtime.inspect(out["name"])
        assert st.gpu_uuids == ["MockMI355X-7"]
        await d.stop()

    run(main())


def test_cpu_apply_lowest_free_sorted(run):
    async def main():
        store = MemoryStore()
        cs = await CpuScheduler.create(store, None, count=8)
        cpuset = cs.apply(3)
        assert cpuset == "0,1,2"
        cpuset2 = cs.apply(2)
        assert cpuset2 == "3,4"
        cs.restore("1,3")
        assert cs.apply(2) == "1,3"
        with pytest.raises(CpuNotEnough):
            cs.apply(9)

    run(main())


def test_port_apply_in_range_and_persist_shape(run):
    async def main():
        store = MemoryStore()
        q = WorkQueue(store)
        q.start()
        ps = await PortScheduler.create(store, q, 40000, 40009)
        ports = ps.apply(4)
        assert len(set(ports)) == 4
        assert all(40000 <= p <= 40009 for p in ports)
        status = ps.get_port_status()
        assert status["AvailableCount"] == 6
        await q.drain()
        kv = await store.get(Resource.PORTS, "usedPortSetKey")
        data = json.loads(kv.value)
        # reference Go shape: UsedPortSet is {"<port>": {}}
        assert set(data["UsedPortSet"].keys()) == {str(p) for p in ports}
        assert data["StartPort"] == 40000 and data["EndPort"] == 40009
        # exhaust the range: dense fallback must fill deterministically
        more = ps.apply(6)
        assert len(more) == 6
        with pytest.raises(PortNotEnough):
            ps.apply(1)
        ps.restore(ports)
        assert ps.get_port_status()["AvailableCount"] == 4
        await q.close()

    run(main())


def test_port_state_reload(run):
    async def main():
        store = MemoryStore()
        ps = await PortScheduler.create(store, None, 40000, 40100)
        got = ps.apply(3)
        await ps.persist()
        ps2 = await PortScheduler.create(store, None, 40000, 40100)
        assert ps2.get_port_status()["AvailableCount"] == 98
        ps2.restore(got)
        assert ps2.get_port_status()["AvailableCount"] == 101

    run(main())


def test_version_map_bump_and_persist(run):
    async def main():
        store = MemoryStore()
        q = WorkQueue(store)
        q.start()
        vm = VersionMap(store, q, "containerVersionMapKey")
        await vm.load()
        assert vm.get("foo") is None
        assert vm.bump("foo") == 1
        assert vm.bump("foo") == 2
        vm.set("bar", 7)
        await q.drain()
        kv = await store.get(Resource.VERSIONS, "containerVersionMapKey")
        assert json.loads(kv.value) == {"foo": 2, "bar": 7}
        vm2 = VersionMap(store, q, "containerVersionMapKey")
        await vm2.load()
        assert vm2.get("foo") == 2
        vm2.remove("foo")
        assert not vm2.exists("foo")
        await q.close()

    run(main())


def test_merge_map(run):
    async def main():
        store = MemoryStore()
        mm = MergeMap(store, None, "containerMergeMapKey")
        await mm.load()
        mm.set("foo-1", "/data/merges/foo/foo-1")
        mm.set("foo-2", "/data/merges/foo/foo-2")
        mm.set("bar-1", "/data/merges/bar/bar-1")
        assert mm.get("foo-1") == "/data/merges/foo/foo-1"
        mm.remove_prefix("foo-")
        assert mm.get("foo-1") is None and mm.get("foo-2") is None
        assert mm.get("bar-1") is not None

    run(main())


def test_probe_cache_file_roundtrip(tmp_path):
    from gpu_docker_api_amd.parallel.topology import Topology
    import json

    p = tmp_path / "probe.json"
    p.write_text(json.dumps({"gpus": ["a", "b"], "p2p_gbps": [[0, 140.0], [141.0, 0]]}))
    probe = Topology.load_probe_file(str(p))
    t = Topology([[0.0, 50.0], [50.0, 0.0]], ["a", "b"])
    t.overlay_measured(probe)
    assert t.bandwidth("a", "b") == 140.0
    # corrupt file degrades to None, not a crash
    p.write_text("{not json")
    assert Topology.load_probe_file(str(p)) is None


def test_cpu_numa_preferred_allocation(run):
    """MI355X extension: cpuset drawn from the GPUs' NUMA node first,
    spilling to other nodes only when the preferred ones run dry."""
    from gpu_docker_api_amd.parallel.cpu import CpuScheduler
    from gpu_docker_api_amd.state.store import MemoryStore

    async def main():
        store = MemoryStore()
        nodes = {c: (0 if c < 4 else 1) for c in range(8)}
        cpu = await CpuScheduler.create(store, None, count=8, cpu_nodes=nodes)
        assert cpu.apply(2, preferred_nodes=[1]) == "4,5"
        # spill: node 1 has only 2 free left -> take them + lowest others
        assert cpu.apply(4, preferred_nodes=[1]) == "0,1,6,7"
        cpu.restore(["0", "1", "4", "5", "6", "7"])
        # no preference -> lowest ids
        assert cpu.apply(3) == "0,1,2"

    run(main())


def test_gpu_numa_nodes_resolution(tmp_path, run):
    from gpu_docker_api_amd.models import ContainerRun
    from helpers import make_daemon

    async def main():
        d = await make_daemon(tmp_path)
        # mock inventory: GPUs 0-3 node 0, GPUs 4-7 node 1
        assert d.replicaset._gpu_numa_nodes(["MockMI355X-1", "MockMI355X-6"]) == [0, 1]
        assert d.replicaset._gpu_numa_nodes(["MockMI355X-2"]) == [0]
        assert d.replicaset._gpu_numa_nodes([]) == []
        await d.stop()

    run(main())
