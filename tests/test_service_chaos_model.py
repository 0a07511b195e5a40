"""Model-checked chaos over the whole control plane: random interleaved
lifecycle operations across tenants, with global invariants verified after
every step — the allocator must always agree with the live specs, versions
must be monotonic, and no operation may leak or double-book resources."""
import random

import pytest

from gpu_docker_api_amd.models import ContainerRun, CpuPatch, GpuPatch, MemoryPatch, PatchRequest
from gpu_docker_api_amd.state.keys import Resource
from gpu_docker_api_amd.xerrors import GdaError
from helpers import make_daemon


async def _invariants(d, tenants):
    # GPUs/CPUs allocated == sum over live, non-released replicaSets
    want_gpu, want_cpu = 0, 0
    for name in tenants:
        v = d.container_versions.get(name)
        if v is None:
            continue
        vname = f"{name}-{v}"
        if vname in d.replicaset._released:
            continue
        kv = await d.store.get_or_none(Resource.CONTAINERS, name)
        assert kv is not None, f"{name}: version map has v{v} but no spec"
        from gpu_docker_api_amd.models.etcd import ContainerSpec

        spec = ContainerSpec.deserialize(kv.value)
        assert spec.version == v, f"{name}: spec v{spec.version} != map v{v}"
        want_gpu += len(spec.gpu_uuids)
        want_cpu += len([c for c in spec.cpuset_cpus.split(",") if c])
    assert sum(d.gpu.get_gpu_status().values()) == want_gpu
    assert sum(d.cpu.get_cpu_status().values()) == want_cpu


def test_service_chaos_model(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        rng = random.Random(20260914)
        tenants = [f"t{i}" for i in range(5)]

        async def step():
            name = rng.choice(tenants)
            op = rng.choice(
                ["run", "patch_gpu", "patch_cpu", "patch_mem", "rollback",
                 "stop", "continue", "restart", "delete"]
            )
            try:
                if op == "run":
                    await d.replicaset.run_gpu_container(
                        ContainerRun(
                            image_name="img",
                            replica_set_name=name,
                            gpu_count=rng.randint(0, 3),
                            cpu_count=rng.randint(0, 3),
                            memory="1GB",
                        )
                    )
                elif op == "patch_gpu":
                    await d.replicaset.patch_container(
                        name, PatchRequest(gpu_patch=GpuPatch(gpu_count=rng.randint(0, 3)))
                    )
                elif op == "patch_cpu":
                    await d.replicaset.patch_container(
                        name, PatchRequest(cpu_patch=CpuPatch(cpu_count=rng.randint(0, 3)))
                    )
                elif op == "patch_mem":
                    await d.replicaset.patch_container(
                        name,
                        PatchRequest(memory_patch=MemoryPatch(memory=f"{rng.randint(1, 4)}GB")),
                    )
                elif op == "rollback":
                    v = d.container_versions.get(name)
                    if v and v > 1:
                        await d.replicaset.rollback_container(name, rng.randint(1, v - 1))
                elif op == "stop":
                    await d.replicaset.stop_container(name)
                elif op == "continue":
                    await d.replicaset.startup_container(name)
                elif op == "restart":
                    await d.replicaset.restart_container(name)
                elif op == "delete":
                    await d.replicaset.delete_container(name)
            except GdaError:
                pass  # expected business failures (exists/not-exist/exhausted)

        for i in range(220):
            await step()
            await _invariants(d, tenants)
        # drain: delete everything; all resources must return
        for name in tenants:
            try:
                await d.replicaset.delete_container(name)
            except GdaError:
                pass
        assert sum(d.gpu.get_gpu_status().values()) == 0
        assert sum(d.cpu.get_cpu_status().values()) == 0
        assert d.ports.get_port_status()["AvailableCount"] == 100
        await d.stop()

    run(main())
