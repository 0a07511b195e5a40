"""Version-history edge cases: delete+recreate lifetimes, rollback across
daemon restarts, deep version chains."""
from gpu_docker_api_amd.models import ContainerRun, GpuPatch, MemoryPatch, PatchRequest
from gpu_docker_api_amd.routers.app import Daemon
from helpers import make_config


def test_delete_recreate_resets_history(tmp_path, run):
    async def main():
        d = Daemon(make_config(tmp_path))
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="old", replica_set_name="app", gpu_count=1)
        )
        await d.replicaset.patch_container("app", PatchRequest(gpu_patch=GpuPatch(gpu_count=2)))
        await d.replicaset.delete_container("app")
        # recreate: history starts fresh (etcd delete ends the key lifetime)
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="new", replica_set_name="app", gpu_count=1)
        )
        hist = await d.replicaset.get_container_history("app")
        assert len(hist) == 1
        assert hist[0]["status"]["config"]["Image"] == "new"
        await d.stop()

    run(main())


def test_rollback_across_daemon_restart(tmp_path, run):
    async def main():
        d = Daemon(make_config(tmp_path))
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="app", gpu_count=1, memory="2GB")
        )
        await d.replicaset.patch_container(
            "app", PatchRequest(memory_patch=MemoryPatch(memory="8GB"))
        )
        await d.stop()

        d2 = Daemon(make_config(tmp_path))
        await d2.start()
        # runtime state (mock) was lost with the process, but spec history
        # survives: rollback re-resolves resources and re-creates v3
        out = await d2.replicaset.rollback_container("app", 1)
        assert out["containerName"] == "app-3"
        st = await d2.runtime.inspect("app-3")
        assert st.memory == 2 * 1024**3
        await d2.stop()

    run(main())


def test_deep_version_chain_history_order(tmp_path, run):
    async def main():
        d = Daemon(make_config(tmp_path))
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="app", memory="1GB")
        )
        for i in range(2, 9):
            await d.replicaset.patch_container(
                "app", PatchRequest(memory_patch=MemoryPatch(memory=f"{i}GB"))
            )
        hist = await d.replicaset.get_container_history("app")
        assert [h["version"] for h in hist] == list(range(8, 0, -1))
        # every entry's stored spec matches its version
        for h in hist:
            assert h["status"]["version"] == h["version"]
        # roll back to the middle, then verify the chain extended
        out = await d.replicaset.rollback_container("app", 4)
        assert out["containerName"] == "app-9"
        st = await d.runtime.inspect("app-9")
        assert st.memory == 4 * 1024**3
        await d.stop()

    run(main())
