"""ReplicaSet + Volume service flows against the mock runtime (real dirs)."""
import os

import pytest

from gpu_docker_api_amd.models import (
    Bind,
    ContainerCommit,
    ContainerExecute,
    ContainerRun,
    CpuPatch,
    GpuPatch,
    MemoryPatch,
    PatchRequest,
    VolumeCreate,
    VolumePatch,
)
from gpu_docker_api_amd.xerrors import (
    ContainerExisted,
    GpuNotEnough,
    NoPatchRequired,
    NoRollbackRequired,
)
from helpers import make_daemon


def _run_req(name="demo", gpus=2, cpus=2, **kw):
    return ContainerRun(
        image_name="ubuntu:22.04",
        replica_set_name=name,
        gpu_count=gpus,
        cpu_count=cpus,
        memory="2GB",
        container_ports=["8080"],
        **kw,
    )


def test_run_creates_versioned_container_with_resources(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        out = await d.replicaset.run_gpu_container(_run_req())
        assert out["name"] == "demo-1"
        st = await d.runtime.inspect("demo-1")
        assert st.running
        assert len(st.gpu_uuids) == 2
        assert st.cpuset_cpus == "0,1"
        assert st.memory == 2 * 1024**3
        assert "CONTAINER_VERSION=1" in st.env
        # one host port bound for the requested container port
        assert list(st.port_bindings) == ["8080/tcp"]
        # schedulers see the allocation
        assert sum(d.gpu.get_gpu_status().values()) == 2
        # spec persisted with byte-compatible top-level shape
        info = await d.replicaset.get_container_info("demo")
        assert set(info) == {
            "version",
            "createTime",
            "config",
            "hostConfig",
            "networkingConfig",
            "platform",
            "containerName",
        }
        assert info["version"] == 1
        with pytest.raises(ContainerExisted):
            await d.replicaset.run_gpu_container(_run_req())
        await d.stop()

    run(main())


def test_run_failure_unwinds_all_resources(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        d.runtime.fail_on.add("start")
        with pytest.raises(RuntimeError):
            await d.replicaset.run_gpu_container(_run_req())
        # saga must have released gpu/cpu/ports, removed the container,
        # and rolled back the version map
        assert sum(d.gpu.get_gpu_status().values()) == 0
        assert sum(d.cpu.get_cpu_status().values()) == 0
        assert d.ports.get_port_status()["AvailableCount"] == 100
        assert await d.runtime.inspect("demo-1") is None
        assert d.container_versions.get("demo") is None
        # after the failure the same name can be used again
        d.runtime.fail_on.clear()
        out = await d.replicaset.run_gpu_container(_run_req())
        assert out["name"] == "demo-1"
        await d.stop()

    run(main())


def test_gpu_not_enough_releases_nothing(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        with pytest.raises(GpuNotEnough):
            await d.replicaset.run_gpu_container(_run_req(gpus=9))
        assert sum(d.gpu.get_gpu_status().values()) == 0
        await d.stop()

    run(main())


def test_patch_gpu_rescale_rolling_replace(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=1))
        # drop a marker file into the writable layer; it must survive replace
        st1 = await d.runtime.inspect("demo-1")
        with open(os.path.join(st1.upper_dir, "marker.txt"), "w") as f:
            f.write("data-v1")
        out = await d.replicaset.patch_container(
            "demo", PatchRequest(gpu_patch=GpuPatch(gpu_count=2))
        )
        assert out["containerName"] == "demo-2"
        assert await d.runtime.inspect("demo-1") is None
        st2 = await d.runtime.inspect("demo-2")
        assert st2.running and len(st2.gpu_uuids) == 2
        # data migrated
        assert open(os.path.join(st2.upper_dir, "marker.txt")).read() == "data-v1"
        # old layer preserved under merges/
        preserved = d.merges.get("demo-1")
        assert preserved and os.path.exists(os.path.join(preserved, "marker.txt"))
        # exactly 2 GPUs allocated now
        assert sum(d.gpu.get_gpu_status().values()) == 2
        # history has both versions, newest first
        hist = await d.replicaset.get_container_history("demo")
        assert [h["version"] for h in hist] == [2, 1]
        await d.stop()

    run(main())


def test_patch_noop_raises_no_need_patch(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=1, cpus=1))
        with pytest.raises(NoPatchRequired):
            await d.replicaset.patch_container(
                "demo",
                PatchRequest(
                    gpu_patch=GpuPatch(gpu_count=1),
                    cpu_patch=CpuPatch(cpu_count=1),
                    memory_patch=MemoryPatch(memory="2GB"),
                ),
            )
        # but the empty patch (recreate-as-is) is allowed
        out = await d.replicaset.patch_container("demo", PatchRequest())
        assert out["containerName"] == "demo-2"
        await d.stop()

    run(main())


def test_patch_failure_unwinds_to_previous_version(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=1))
        d.runtime.fail_on.add("start")
        with pytest.raises(RuntimeError):
            await d.replicaset.patch_container(
                "demo", PatchRequest(gpu_patch=GpuPatch(gpu_count=2))
            )
        d.runtime.fail_on.clear()
        # old container still there, still version 1, exactly 1 GPU used
        assert d.container_versions.get("demo") == 1
        st = await d.runtime.inspect("demo-1")
        assert st is not None and st.running
        assert sum(d.gpu.get_gpu_status().values()) == 1
        assert await d.runtime.inspect("demo-2") is None
        await d.stop()

    run(main())


def test_rollback_restores_old_spec(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=1))
        await d.replicaset.patch_container(
            "demo", PatchRequest(gpu_patch=GpuPatch(gpu_count=3))
        )
        st2 = await d.runtime.inspect("demo-2")
        assert len(st2.gpu_uuids) == 3
        with pytest.raises(NoRollbackRequired):
            await d.replicaset.rollback_container("demo", 2)
        out = await d.replicaset.rollback_container("demo", 1)
        assert out["containerName"] == "demo-3"
        st3 = await d.runtime.inspect("demo-3")
        # resources re-resolved to version-1 shape: 1 GPU
        assert len(st3.gpu_uuids) == 1
        assert sum(d.gpu.get_gpu_status().values()) == 1
        # memory restored exactly (the reference inflates 1024x here)
        assert st3.memory == 2 * 1024**3
        await d.stop()

    run(main())


def test_rollback_with_data_restore(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=0, cpus=0))
        st1 = await d.runtime.inspect("demo-1")
        with open(os.path.join(st1.upper_dir, "gen.txt"), "w") as f:
            f.write("generation-1")
        await d.replicaset.patch_container("demo", PatchRequest())  # -> demo-2
        st2 = await d.runtime.inspect("demo-2")
        with open(os.path.join(st2.upper_dir, "gen.txt"), "w") as f:
            f.write("generation-2")
        # plain rollback keeps CURRENT data (reference behavior)
        out = await d.replicaset.rollback_container("demo", 1)
        st3 = await d.runtime.inspect(out["containerName"])
        assert open(os.path.join(st3.upper_dir, "gen.txt")).read() == "generation-2"
        # rollback with restore_data brings back version 1's layer
        out = await d.replicaset.rollback_container("demo", 1, restore_data=True)
        st4 = await d.runtime.inspect(out["containerName"])
        assert open(os.path.join(st4.upper_dir, "gen.txt")).read() == "generation-1"
        await d.stop()

    run(main())


def test_merge_layer_retention_pruning(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        d.replicaset.cfg.keep_merge_layers = 2
        await d.replicaset.run_gpu_container(_run_req(gpus=0, cpus=0))
        for _ in range(5):
            await d.replicaset.patch_container("demo", PatchRequest())
        snap = d.replicaset.merges.snapshot()
        kept = sorted(k for k in snap if k.startswith("demo-"))
        assert kept == ["demo-4", "demo-5"]
        # pruned layer dirs are gone from disk
        assert not os.path.exists(os.path.join(d.cfg.merges_dir, "demo", "demo-1"))
        assert os.path.exists(snap["demo-5"])
        await d.stop()

    run(main())


def test_stop_releases_and_startup_reacquires(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=2, cpus=2))
        await d.replicaset.stop_container("demo")
        assert sum(d.gpu.get_gpu_status().values()) == 0
        assert sum(d.cpu.get_cpu_status().values()) == 0
        assert d.ports.get_port_status()["AvailableCount"] == 100
        st = await d.runtime.inspect("demo-1")
        assert not st.running
        await d.replicaset.startup_container("demo")
        assert sum(d.gpu.get_gpu_status().values()) == 2
        st = await d.runtime.inspect("demo-1")
        assert st.running
        # double stop is safe
        await d.replicaset.stop_container("demo")
        await d.replicaset.stop_container("demo")
        assert sum(d.gpu.get_gpu_status().values()) == 0
        await d.stop()

    run(main())


def test_patch_of_stopped_container_reacquires_resources(tmp_path, run):
    """A stopped replicaSet released its GPUs; patching it must re-acquire
    them (or fail) — otherwise the replacement double-books."""

    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(name="a", gpus=2))
        await d.replicaset.stop_container("a")
        assert sum(d.gpu.get_gpu_status().values()) == 0
        # another tenant takes 7 of 8 GPUs
        await d.replicaset.run_gpu_container(_run_req(name="b", gpus=7))
        # patch a: its old 2 GPUs can't all be re-acquired -> must fail clean
        with pytest.raises(GpuNotEnough):
            await d.replicaset.patch_container(
                "a", PatchRequest(gpu_patch=GpuPatch(gpu_count=2))
            )
        assert sum(d.gpu.get_gpu_status().values()) == 7  # unchanged
        await d.replicaset.delete_container("b")
        # now it works: re-acquire + rescale
        out = await d.replicaset.patch_container(
            "a", PatchRequest(gpu_patch=GpuPatch(gpu_count=3))
        )
        st = await d.runtime.inspect(out["containerName"])
        assert len(st.gpu_uuids) == 3
        assert sum(d.gpu.get_gpu_status().values()) == 3
        await d.stop()

    run(main())


def test_pause_keeps_resources(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=2))
        await d.replicaset.pause_container("demo")
        assert (await d.runtime.inspect("demo-1")).paused
        assert sum(d.gpu.get_gpu_status().values()) == 2
        await d.replicaset.startup_container("demo")
        assert not (await d.runtime.inspect("demo-1")).paused
        await d.stop()

    run(main())


def test_restart_rolls_replacement(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=2))
        out = await d.replicaset.restart_container("demo")
        assert out["containerName"] == "demo-2"
        assert await d.runtime.inspect("demo-1") is None
        assert sum(d.gpu.get_gpu_status().values()) == 2
        await d.stop()

    run(main())


def test_delete_releases_everything(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req())
        await d.replicaset.patch_container("demo", PatchRequest(gpu_patch=GpuPatch(gpu_count=3)))
        await d.replicaset.delete_container("demo")
        assert sum(d.gpu.get_gpu_status().values()) == 0
        assert d.container_versions.get("demo") is None
        assert await d.runtime.inspect("demo-2") is None
        assert d.merges.get("demo-1") is None
        if d.queue:
            await d.queue.drain()
        from gpu_docker_api_amd.state.keys import Resource

        assert await d.store.get_or_none(Resource.CONTAINERS, "demo") is None
        # name is reusable, version restarts at 1
        out = await d.replicaset.run_gpu_container(_run_req())
        assert out["name"] == "demo-1"
        await d.stop()

    run(main())


def test_execute_and_commit(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(gpus=0, cpus=0))
        out, rc = await d.replicaset.execute_container(
            "demo", ContainerExecute(cmd=["sh", "-c", "echo hello-$PWD"])
        )
        assert "hello-" in out and rc == 0
        _out, rc = await d.replicaset.execute_container(
            "demo", ContainerExecute(cmd=["sh", "-c", "exit 3"])
        )
        assert rc == 3
        image = await d.replicaset.commit_container(
            "demo", ContainerCommit(new_image_name="demo-img")
        )
        assert image == "demo-img"
        assert "demo-img" in d.runtime.images
        await d.stop()

    run(main())


def test_volume_create_resize_delete_with_data(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        out = await d.volume.create_volume(VolumeCreate(name="vol", size="10GB"))
        assert out["name"] == "vol-1"
        vs = await d.runtime.volume_inspect("vol-1")
        with open(os.path.join(vs.mountpoint, "payload.bin"), "wb") as f:
            f.write(b"x" * 4096)
        out2 = await d.volume.patch_volume_size("vol", "50GB")
        assert out2["name"] == "vol-2"
        # data migrated, old volume actually deleted (reference leaks it)
        vs2 = await d.runtime.volume_inspect("vol-2")
        assert os.path.exists(os.path.join(vs2.mountpoint, "payload.bin"))
        assert await d.runtime.volume_inspect("vol-1") is None
        # same-size patch (different unit, same bytes) => no-need-patch
        with pytest.raises(NoPatchRequired):
            await d.volume.patch_volume_size("vol", "51200MB")
        hist = await d.volume.get_volume_history("vol")
        assert [h["version"] for h in hist] == [2, 1]
        await d.volume.delete_volume("vol")
        assert d.volume_versions.get("vol") is None
        await d.stop()

    run(main())


def test_volume_shrink_guard(tmp_path, run):
    async def main():
        from gpu_docker_api_amd.xerrors import VolumeSizeUsedGreaterThanReduced

        d = await make_daemon(tmp_path)
        await d.volume.create_volume(VolumeCreate(name="vol", size="10GB"))
        vs = await d.runtime.volume_inspect("vol-1")
        with open(os.path.join(vs.mountpoint, "big.bin"), "wb") as f:
            f.write(b"y" * (2 * 1024 * 1024))  # 2 MB used
        with pytest.raises(VolumeSizeUsedGreaterThanReduced):
            await d.volume.patch_volume_size("vol", "1MB")
        await d.stop()

    run(main())


def test_container_patch_volume_bind(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.volume.create_volume(VolumeCreate(name="vol", size="10GB"))
        req = _run_req(gpus=0, cpus=0)
        req.binds = [Bind(src="vol-1", dest="/data")]
        await d.replicaset.run_gpu_container(req)
        out = await d.replicaset.patch_container(
            "demo",
            PatchRequest(
                volume_patch=VolumePatch(
                    old_bind=Bind(src="vol-1", dest="/data"),
                    new_bind=Bind(src="vol-2", dest="/data"),
                )
            ),
        )
        st = await d.runtime.inspect(out["containerName"])
        assert st.binds == ["vol-2:/data"]
        await d.stop()

    run(main())


def test_container_logs_endpoint(tmp_path, run):
    """Logs extension: the proc runtime captures console output and the
    service/route surface it with tail semantics."""
    import asyncio

    from gpu_docker_api_amd.models import ContainerRun

    async def main():
        from helpers import make_config
        from gpu_docker_api_amd.routers.app import Daemon

        cfg = make_config(tmp_path, runtime="proc")
        d = Daemon(cfg)
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(
                image_name="img",
                replica_set_name="logs",
                gpu_count=0,
                cmd=["sh", "-c", "echo line-one; echo line-two; sleep 30"],
            )
        )
        out = ""
        for _ in range(100):  # console.log is written asynchronously
            out = await d.replicaset.get_container_logs("logs")
            if "line-two" in out:
                break
            await asyncio.sleep(0.05)
        assert "line-one" in out and "line-two" in out
        # tail=1 returns only the newest line
        tail1 = await d.replicaset.get_container_logs("logs", tail=1)
        assert "line-two" in tail1 and "line-one" not in tail1
        await d.replicaset.delete_container("logs")
        await d.stop()

    run(main())


def test_container_stats_endpoint(tmp_path, run):
    """Stats extension: live cpu/memory of the current version (proc
    runtime reads cgroup v2 or /proc)."""
    import asyncio

    from gpu_docker_api_amd.models import ContainerRun

    async def main():
        from helpers import make_config
        from gpu_docker_api_amd.routers.app import Daemon

        cfg = make_config(tmp_path, runtime="proc")
        d = Daemon(cfg)
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="st", gpu_count=0,
                         cmd=["sleep", "30"])
        )
        st = await d.replicaset.get_container_stats("st")
        assert st["running"] is True
        assert st["memoryBytes"] > 0
        assert st["pids"] >= 1
        await d.replicaset.delete_container("st")
        await d.stop()

    run(main())


def test_container_stats_includes_gpu_hbm(tmp_path, run):
    from gpu_docker_api_amd.models import ContainerRun

    async def main():
        from helpers import make_daemon

        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="g", gpu_count=2)
        )
        st = await d.replicaset.get_container_stats("g")
        assert len(st["gpus"]) == 2
        for g in st["gpus"]:
            assert g["uuid"].startswith("MockMI355X-")
            assert g["hbmTotalBytes"] and g["hbmTotalBytes"] > 200 * 1024**3
        await d.replicaset.delete_container("g")
        await d.stop()

    run(main())


def test_cpuset_mems_recorded_with_gpu_nodes(tmp_path, run):
    from gpu_docker_api_amd.models import ContainerRun

    async def main():
        from helpers import make_daemon

        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="mm", gpu_count=2,
                         cpu_count=2)
        )
        spec = await d.replicaset._load_spec("mm")
        nodes = d.replicaset._gpu_numa_nodes(spec.gpu_uuids)
        assert spec.host_config.get("CpusetMems") == ",".join(map(str, nodes))
        await d.replicaset.delete_container("mm")
        await d.stop()

    run(main())
