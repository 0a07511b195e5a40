"""Concurrency (the reference's version maps race under concurrent handlers,
SURVEY.md §5.2) and daemon-restart recovery (state reload from WAL)."""
import asyncio
import os

from gpu_docker_api_amd.models import ContainerRun, GpuPatch, PatchRequest, VolumeCreate
from gpu_docker_api_amd.routers.app import Daemon
from helpers import make_config


def test_concurrent_runs_allocate_disjoint_resources(tmp_path, run):
    """8 concurrent 1-GPU runs: exactly 8 distinct GPUs, 8 distinct cpusets,
    no double-allocation, no lost version-map updates."""

    async def main():
        d = Daemon(make_config(tmp_path))
        await d.start()

        async def one(i):
            return await d.replicaset.run_gpu_container(
                ContainerRun(
                    image_name="img",
                    replica_set_name=f"c{i}",
                    gpu_count=1,
                    cpu_count=1,
                    container_ports=["80"],
                )
            )

        results = await asyncio.gather(*[one(i) for i in range(8)])
        assert sorted(r["name"] for r in results) == sorted(f"c{i}-1" for i in range(8))
        # every GPU allocated exactly once
        assert sum(d.gpu.get_gpu_status().values()) == 8
        states = [await d.runtime.inspect(f"c{i}-1") for i in range(8)]
        gpus = [u for s in states for u in s.gpu_uuids]
        assert len(set(gpus)) == 8
        cpusets = [s.cpuset_cpus for s in states]
        assert len(set(cpusets)) == 8
        ports = [b[0]["HostPort"] for s in states for b in s.port_bindings.values()]
        assert len(set(ports)) == 8
        await d.stop()

    run(main())


def test_concurrent_patches_on_distinct_sets(tmp_path, run):
    async def main():
        d = Daemon(make_config(tmp_path))
        await d.start()
        for i in range(4):
            await d.replicaset.run_gpu_container(
                ContainerRun(image_name="img", replica_set_name=f"c{i}", gpu_count=1)
            )

        async def patch(i):
            return await d.replicaset.patch_container(
                f"c{i}", PatchRequest(gpu_patch=GpuPatch(gpu_count=2))
            )

        results = await asyncio.gather(*[patch(i) for i in range(4)])
        assert sorted(r["containerName"] for r in results) == sorted(
            f"c{i}-2" for i in range(4)
        )
        assert sum(d.gpu.get_gpu_status().values()) == 8
        await d.stop()

    run(main())


def test_daemon_restart_recovers_everything(tmp_path, run):
    """Kill the daemon (without graceful persist beyond normal write-behind)
    and restart over the same data dir: allocations, versions, history and
    running containers' specs must all come back."""

    async def main():
        cfg = make_config(tmp_path)
        d = Daemon(cfg)
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="app", gpu_count=2, cpu_count=2)
        )
        await d.replicaset.patch_container("app", PatchRequest(gpu_patch=GpuPatch(gpu_count=3)))
        await d.volume.create_volume(VolumeCreate(name="vol", size="5GB"))
        await d.queue.drain()
        # no graceful stop: simulate crash by just abandoning the instance
        await d.store.close()

        d2 = Daemon(make_config(tmp_path))
        await d2.start()
        # version map recovered
        assert d2.container_versions.get("app") == 2
        assert d2.volume_versions.get("vol") == 1
        # scheduler state recovered: 3 GPUs + 2 CPUs still allocated
        assert sum(d2.gpu.get_gpu_status().values()) == 3
        assert sum(d2.cpu.get_cpu_status().values()) == 2
        # history intact across restart (WAL replay keeps MVCC revisions)
        hist = await d2.replicaset.get_container_history("app")
        assert [h["version"] for h in hist] == [2, 1]
        info = await d2.replicaset.get_container_info("app")
        assert info["containerName"] == "app-2"
        await d2.stop()

    run(main())


def test_graceful_shutdown_persists_sync(tmp_path, run):
    async def main():
        cfg = make_config(tmp_path)
        d = Daemon(cfg)
        await d.start()
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="app", gpu_count=1)
        )
        await d.stop()  # graceful: drains queue + sync persists

        d2 = Daemon(make_config(tmp_path))
        await d2.start()
        assert sum(d2.gpu.get_gpu_status().values()) == 1
        assert d2.container_versions.get("app") == 1
        await d2.stop()

    run(main())
