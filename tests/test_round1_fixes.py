"""Regression tests for the round-1 review findings (VERDICT.md / ADVICE.md).

1. Path traversal via resource names: names like '..'
   must never reach a filesystem join.
2. Released-resource persistence (VERDICT weak #1 / ADVICE #3): a daemon
   restart must not forget that a stopped container released its GPUs.
3. Spec-persist saga compensation: a failed replace must
   restore the CONTAINERS key, and rollback-by-version must stay correct
   even with compensated writes in the key's history.
"""
import pytest

from gpu_docker_api_amd.models import ContainerRun, MemoryPatch, CpuPatch, GpuPatch, PatchRequest
from gpu_docker_api_amd.routers.app import Daemon
from gpu_docker_api_amd.utils.names import safe_subpath, valid_name
from gpu_docker_api_amd.xerrors import GpuNotEnough
from helpers import make_config, make_daemon


def _run_req(name="demo", gpus=2, cpus=2, **kw):
    return ContainerRun(
        image_name="ubuntu:22.04",
        replica_set_name=name,
        gpu_count=gpus,
        cpu_count=cpus,
        memory="2GB",
        **kw,
    )


# --------------------------------------------------------------- name safety
def test_valid_name_grammar():
    for ok in ("a", "web_1", "A.b.c", "x" * 64, "_hidden", "9lives"):
        assert valid_name(ok), ok
    for bad in ("", ".", "..", "...", "a/b", "a-b", "/abs", "a b", "a\n", "x" * 65, "-x", ".x"):
        assert not valid_name(bad), bad


def test_safe_subpath_blocks_escape(tmp_path):
    base = tmp_path / "data"
    base.mkdir()
    assert safe_subpath(str(base), "a", "a-1").endswith("a/a-1")
    for parts in (("..",), ("a", ".."), ("../x",), ("/abs",), (".",), ("",)):
        with pytest.raises(ValueError):
            safe_subpath(str(base), *parts)


def test_api_rejects_traversal_names(tmp_path, run):
    from fastapi.testclient import TestClient

    from gpu_docker_api_amd.routers.app import build_app

    app = build_app(make_config(tmp_path))
    with TestClient(app) as client:
        # a sentinel file directly in data_dir: the round-1 advisor's repro
        # deleted it via DELETE /replicaSet/..
        sentinel = tmp_path / "state" / "sentinel"
        sentinel.write_text("keep me")
        body = {"imageName": "ubuntu:22.04", "replicaSetName": "..", "gpuCount": 0}
        r = client.post("/api/v1/replicaSet", json=body).json()
        assert r["code"] != 200
        # the HTTP client normalizes a literal '..' segment away, so the
        # hostile path arrives percent-encoded; starlette decodes it into
        # the route's {name} param
        resp = client.delete("/api/v1/replicaSet/%2e%2e")
        assert resp.status_code == 404 or resp.json()["code"] != 200
        assert sentinel.read_text() == "keep me"
        # volume side
        r = client.post("/api/v1/volumes", json={"name": "..", "size": "1GB"}).json()
        assert r["code"] != 200
        r = client.post("/api/v1/volumes", json={"name": "a/../..", "size": "1GB"}).json()
        assert r["code"] != 200


def test_proc_runtime_refuses_unsafe_names(tmp_path, run):
    from gpu_docker_api_amd.models.etcd import ContainerSpec
    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path / "rt"), use_cgroups=False)
        spec = ContainerSpec()
        spec.container_name = "../../escape"
        with pytest.raises(ValueError):
            await rt.create(spec)
        with pytest.raises(ValueError):
            await rt.volume_create("..")
        await rt.close()

    run(main())


# ------------------------------------------------- released-state persistence
def test_released_set_survives_daemon_restart(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(name="a", gpus=2))
        await d.replicaset.stop_container("a")
        assert "a-1" in d.replicaset._released
        assert sum(d.gpu.get_gpu_status().values()) == 0  # released
        await d.queue.close()  # flush write-behind state

        # daemon restart on the same store (crash: no graceful d.stop())
        d2 = Daemon(make_config(tmp_path), store=d.store)
        await d2.start()
        assert "a-1" in d2.replicaset._released

        # another tenant takes 7 of the 8 GPUs -> at least one of a's
        # formerly-released GPUs is now owned by b
        await d2.replicaset.run_gpu_container(_run_req(name="b", gpus=7))
        with pytest.raises(GpuNotEnough):
            await d2.replicaset.startup_container("a")
        # the failed startup must not leak allocations: still exactly b's 7
        assert sum(d2.gpu.get_gpu_status().values()) == 7
        await d2.stop()

    run(main())


def test_patch_after_restart_reacquires_released(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(name="a", gpus=2))
        await d.replicaset.stop_container("a")
        await d.queue.close()

        d2 = Daemon(make_config(tmp_path), store=d.store)
        await d2.start()
        out = await d2.replicaset.patch_container(
            "a", PatchRequest(gpu_patch=GpuPatch(gpu_count=1))
        )
        assert out["containerName"] == "a-2"
        # re-acquired then rescaled: exactly one GPU allocated, not zero
        # (double-booking would show 1 here too but with b absent the real
        # signal is the released set being consulted at all — covered above)
        assert sum(d2.gpu.get_gpu_status().values()) == 1
        assert "a-1" not in d2.replicaset._released
        await d2.stop()

    run(main())


# ------------------------------------------------ spec compensation + rollback
def test_failed_replace_restores_spec_and_rollback_alignment(tmp_path, run):
    async def main():
        d = await make_daemon(tmp_path)
        await d.replicaset.run_gpu_container(_run_req(name="a", gpus=1))

        # failed patch: start of the replacement container blows up
        d.runtime.fail_on.add("start")
        with pytest.raises(RuntimeError):
            await d.replicaset.patch_container(
                "a", PatchRequest(memory_patch=MemoryPatch(memory="8GB"))
            )
        d.runtime.fail_on.clear()

        # the store must agree with the reverted version map
        spec = await d.replicaset._load_spec("a")
        assert spec.version == 1
        assert spec.memory_bytes == 2 * 1024**3  # not the failed 8GB
        assert d.container_versions.get("a") == 1

        # real v2 (memory 4GB) and v3 (cpu change) succeed
        await d.replicaset.patch_container(
            "a", PatchRequest(memory_patch=MemoryPatch(memory="4GB"))
        )
        await d.replicaset.patch_container(
            "a", PatchRequest(cpu_patch=CpuPatch(cpu_count=1))
        )
        assert d.container_versions.get("a") == 3

        # rollback to version 2 must restore the REAL v2 spec (4GB), not a
        # failed-write artifact that shares its per-key version counter
        await d.replicaset.rollback_container("a", 2)
        spec = await d.replicaset._load_spec("a")
        assert spec.memory_bytes == 4 * 1024**3
        await d.stop()

    run(main())


# ------------------------------------------------------- property fuzzing
def test_safe_subpath_property_fuzz(tmp_path):
    """For ANY string (hypothesis), safe_subpath either raises ValueError or
    returns a path strictly below base after symlink/dot resolution."""
    import os

    from hypothesis import given, settings, strategies as st

    from gpu_docker_api_amd.utils.names import safe_subpath

    base = str(tmp_path / "base")
    os.makedirs(base, exist_ok=True)
    base_real = os.path.realpath(base)

    @settings(max_examples=300, deadline=None)
    @given(st.text(min_size=0, max_size=64))
    def check(name):
        try:
            p = safe_subpath(base, name)
        except ValueError:
            return
        assert os.path.realpath(p).startswith(base_real + os.sep)

    check()


def test_valid_name_implies_safe_subpath(tmp_path):
    """Every name the routers accept must be safe to join."""
    import os

    from hypothesis import given, settings, strategies as st

    from gpu_docker_api_amd.utils.names import safe_subpath, valid_name

    base = str(tmp_path / "b")
    os.makedirs(base, exist_ok=True)

    @settings(max_examples=300, deadline=None)
    @given(st.text(min_size=1, max_size=64))
    def check(name):
        if valid_name(name):
            p = safe_subpath(base, name)  # must not raise
            assert os.path.basename(p) == name

    check()


def test_failed_volume_resize_restores_spec(tmp_path, run):
    """Same compensation class as the container side: a resize whose
    migration fails must leave the VOLUMES spec at the previous version,
    agreeing with the reverted version map."""
    from gpu_docker_api_amd.models import VolumeCreate

    async def main():
        d = await make_daemon(tmp_path)
        await d.volume.create_volume(VolumeCreate(name="v", size="1GB"))
        assert d.volume_versions.get("v") == 1

        orig = d.replicaset.copy.move_contents

        async def boom(src, dest):
            raise RuntimeError("injected migrate failure")

        d.volume.copy.move_contents = boom
        with pytest.raises(RuntimeError):
            await d.volume.patch_volume_size("v", "2GB")
        d.volume.copy.move_contents = orig

        assert d.volume_versions.get("v") == 1  # version map reverted
        spec = await d.volume._load_spec("v")
        assert spec.version == 1 and spec.size == "1GB"  # spec agrees
        # and a later resize still works cleanly
        out = await d.volume.patch_volume_size("v", "2GB")
        assert out["name"] == "v-2"
        await d.stop()

    run(main())
