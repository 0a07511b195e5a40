"""GPU-gated tests (run on a real MI355X via gpurun / the round-end driver).

Every test here exercises the native HIP path; ops fail loudly when the
extension is missing, so a silent eager fallback cannot pass these.
"""
import json
import os
import subprocess

import pytest

from conftest import require_gpu

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    require_gpu()
    from gpu_docker_api_amd.ops import hipcore

    return hipcore.load_ext()


def test_device_info(ext):
    n = ext.device_count()
    assert n >= 1
    info = ext.device_info(0)
    assert "gfx95" in info["gcnArchName"], info
    # MI355X: 288 GB HBM3E
    assert info["totalGlobalMem"] > 200 * 1024**3
    assert info["multiProcessorCount"] >= 200


def test_copy_kernel_numerics(ext):
    import torch

    torch.manual_seed(1)
    for n in (16, 1024, 1 << 20, (1 << 20) + 3):  # incl. non-multiple-of-4
        src = torch.randn(n, device="cuda", dtype=torch.float32)
        dst = torch.zeros_like(src)
        ext.copy_f32(dst, src)
        torch.cuda.synchronize()
        assert torch.equal(dst, src), f"copy mismatch at n={n}"


def test_mfma_f32_exact_vs_torch(ext):
    import torch

    torch.manual_seed(2)
    # asymmetric B catches transposed C-writes (CDNA4 guide §3)
    A = torch.randn(128, 256, device="cuda", dtype=torch.float32)
    B = torch.randn(256, 64, device="cuda", dtype=torch.float32)
    C = ext.mfma_f32_matmul(A, B)
    ref = A @ B
    err = (C - ref).abs().max().item()
    rel = err / ref.abs().max().item()
    assert rel < 1e-5, f"mfma f32 rel err {rel}"


def test_mfma_bf16_vs_fp32_reference(ext):
    import torch

    torch.manual_seed(3)
    A = torch.randn(128, 128, device="cuda").bfloat16()
    B = torch.randn(128, 64, device="cuda").bfloat16()
    C = ext.mfma_bf16_matmul(A, B)
    ref = A.float() @ B.float()
    # bf16 inputs, fp32 accumulate: error bounded by bf16 rounding of inputs
    err = (C - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err / scale < 0.05, f"mfma bf16 rel err {err / scale}"


def test_gemm_bf16_numerics_vs_torch(ext):
    import torch

    torch.manual_seed(5)
    # asymmetric operands catch transposed writes (guide §3 / §5.4 rule 16)
    A = (torch.randn(256, 512, device="cuda") * 0.5).bfloat16()
    Bt = (torch.randn(384, 512, device="cuda") * 0.5).bfloat16()
    C = ext.gemm_bf16_bt(A, Bt)
    ref = A.float() @ Bt.float().T
    err = (C - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    assert err / scale < 0.02, f"gemm rel err {err / scale}"


def test_gemm_bf16_throughput_floor(ext):
    # ratcheted regression floor: round-1/2 measured
    # 800-839 TF at 4096^3 on this structure; 75% of the low measurement
    tflops = ext.gemm_bf16_tflops(0, 4096, 10)
    print(f"bf16 GEMM (128^2): {tflops:.0f} TFLOPS @4096^3")
    assert tflops > 600, f"bf16 GEMM regressed: {tflops} TF (floor 600 = 75% of measured 800)"


def test_gemm_bf16_8phase_numerics_and_throughput(ext):
    import torch

    torch.manual_seed(7)
    A = (torch.randn(512, 512, device="cuda") * 0.5).bfloat16()
    Bt = (torch.randn(256, 512, device="cuda") * 0.5).bfloat16()
    ref = A.float() @ Bt.float().T
    # repeat: races in the counted-vmcnt pipeline show as nondeterminism
    for _ in range(3):
        C = ext.gemm_bf16_8ph(A, Bt)
        err = (C - ref).abs().max().item()
        assert err / (ref.abs().max().item() + 1e-6) < 0.02, err
    tflops = ext.gemm_bf16_8ph_tflops(0, 4096, 8)
    print(f"bf16 GEMM (8-phase 256^2): {tflops:.0f} TFLOPS @4096^3")
    # round-2 cold-clock measurements bottom out ~1017 TF; 75% floor
    assert tflops > 760, f"8-phase GEMM regressed: {tflops} TF (floor 760 = 75% of measured 1017)"


def test_validate_gpus_report():
    require_gpu()
    from gpu_docker_api_amd.ops import hipcore

    report = hipcore.validate_gpus(size=2048, iters=3)
    assert report["gpus"], report
    g = report["gpus"][0]
    assert g["hbm_gbps"] > 2000
    assert g["bf16_tflops"] > 100
    assert g["healthy"] is True


def test_hbm_stream_bandwidth_floor(ext):
    # MI355X HBM3E: 8 TB/s peak, ~6.3 achievable; a vectorized grid-stride
    # copy must clear 2 TB/s easily — below that the kernel is broken
    bw = ext.stream_bandwidth_gbps(0, 1024, 10)
    print(f"stream bandwidth: {bw:.0f} GB/s")
    # measured 5092 GB/s in round 1; 75% ratchet
    assert bw > 3800, f"HBM copy bandwidth regressed: {bw} GB/s (floor 3800 = 75% of measured 5092)"


def test_probe_output_shape():
    require_gpu()
    from gpu_docker_api_amd.ops import hipcore

    probe = hipcore.run_probe(mib=256, iters=3)
    n = len(probe["gpus"])
    assert n >= 1
    assert len(probe["hbm_gbps"]) == n
    assert len(probe["p2p_gbps"]) == n
    assert all(len(row) == n for row in probe["p2p_gbps"])
    assert probe["hbm_gbps"][0] > 2000
    if n > 1:
        # xGMI p2p: nonzero off-diagonal
        assert probe["p2p_gbps"][0][1] > 10


def test_rccl_smoke_binary():
    require_gpu()
    from gpu_docker_api_amd.ops import hipcore

    res = hipcore.run_rccl_smoke(mib=16)
    assert res.get("ok"), res
    assert res["world"] >= 1


def test_native_iocopy_loaded_on_gpu_box(tmp_path):
    """The io_uring engine must be present and active on GPU boxes (no
    silent tar fallback in production paths)."""
    require_gpu()
    from gpu_docker_api_amd.ops import iocopy

    assert iocopy.uring_available(), "io_uring unavailable on the GPU box"
    src = tmp_path / "src"
    src.mkdir()
    (src / "f.bin").write_bytes(b"x" * 123456)
    stats = iocopy.copy_tree(str(src), str(tmp_path / "dst"))
    assert stats["io_uring"] is True
    assert (tmp_path / "dst" / "f.bin").read_bytes() == b"x" * 123456


def test_amdsmi_inventory_real():
    require_gpu()
    from gpu_docker_api_amd.parallel.inventory import AmdSmiInventory

    inv = AmdSmiInventory()
    gpus = inv.enumerate()
    assert len(gpus) >= 1
    g = gpus[0]
    assert g.uuid
    assert g.vram_total > 200 * 1024**3
    mat = inv.link_matrix()
    assert len(mat) == len(gpus)


def test_proc_runtime_gpu_visibility(tmp_path, run):
    """End-to-end: a 1-GPU replicaSet on the proc runtime must see exactly
    its allocated GPU via ROCR_VISIBLE_DEVICES."""
    require_gpu()
    import sys

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from helpers import make_daemon

    from gpu_docker_api_amd.models import ContainerRun

    async def main():
        d = await make_daemon(tmp_path, runtime="proc", inventory="amdsmi", copy_engine="auto")
        out = await d.replicaset.run_gpu_container(
            ContainerRun(
                image_name="synthetic:test",
                replica_set_name="gputest",
                gpu_count=1,
                cpu_count=1,
            )
        )
        vis = await d.runtime.execute(
            out["name"], ["sh", "-c", "echo VIS=$ROCR_VISIBLE_DEVICES"]
        )
        assert "VIS=" in vis
        idx = vis.strip().split("=", 1)[1]
        assert idx != "", "GPU index not injected"
        await d.replicaset.delete_container("gputest")
        await d.stop()

    run(main())


def test_proc_runtime_cgroup_limits(tmp_path, run):
    """On the GPU box (root, cgroup v2) the proc runtime must actually
    apply memory.max / cpuset.cpus."""
    require_gpu()
    import sys

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    if not os.path.exists("/sys/fs/cgroup/cgroup.controllers"):
        pytest.skip("no cgroup v2")
    from gpu_docker_api_amd.models.etcd import ContainerSpec
    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=True)
        spec = ContainerSpec()
        spec.config = {"Cmd": ["sleep", "30"]}
        spec.host_config = {}
        spec.container_name = "cg-1"
        spec.memory_bytes = 512 * 1024 * 1024
        await rt.create(spec)
        await rt.start("cg-1")
        p = rt._procs["cg-1"]
        if p.cgroup is None:
            pytest.skip("cgroup writes not permitted in this container")
        with open(os.path.join(p.cgroup, "memory.max")) as f:
            assert f.read().strip() == str(512 * 1024 * 1024)
        with open(os.path.join(p.cgroup, "cgroup.procs")) as f:
            assert str(p.state.pid) in f.read().split()
        await rt.close()

    run(main())


def test_proc_volume_loop_quota_enforced_or_honest(tmp_path, run):
    """Sized volumes: where loop mounts are permitted, writing past the
    size must fail with ENOSPC (real enforcement). Where they are not —
    the round-1 GPU lease containers lack /dev/loop-control and deny
    mount(2) outright (gpurun_out/call1.log: mount -o loop -> EPERM as
    root) — the runtime must RECORD the advisory degradation and the
    volume API must surface it, never silently claim enforcement.
    This replaces the round-1 skip: the test now
    runs and asserts on every environment."""
    require_gpu()
    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False, loop_volumes=True)
        vs = await rt.volume_create("q_1", {"size": "64MB"})
        try:
            if vs.options.get("enforced") == "loop":
                with pytest.raises(OSError):
                    with open(os.path.join(vs.mountpoint, "big.bin"), "wb") as f:
                        f.write(b"x" * (128 * 1024 * 1024))  # 2x the quota
                        f.flush()
                        os.fsync(f.fileno())
                # within quota still works
                with open(os.path.join(vs.mountpoint, "ok.bin"), "wb") as f:
                    f.write(b"y" * (4 * 1024 * 1024))
            else:
                # honesty path: degradation recorded, volume usable
                assert vs.options.get("enforced") == "none", vs.options
                persisted = json.load(open(os.path.join(tmp_path, "volumes", "q_1", "opts.json")))
                assert persisted.get("enforced") == "none"
                with open(os.path.join(vs.mountpoint, "ok.bin"), "wb") as f:
                    f.write(b"y" * (4 * 1024 * 1024))
        finally:
            await rt.volume_remove("q_1")

    run(main())


def test_bench_short_run():
    """bench.py must emit its JSON line on one GPU within minutes."""
    require_gpu()
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        ["python", os.path.join(root, "bench.py"), "--gpus", "1", "--steps", "5", "--warmup", "2"],
        capture_output=True,
        text=True,
        timeout=600,
        cwd=root,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    result = json.loads(line)
    assert result["n_gpus"] == 1
    assert result["value"] > 0
    assert result["higher_is_better"] is False


def test_gemm_fp8_mx_numerics_and_throughput(ext):
    """MX-fp8 (block-scaled mfma 16x16x128, scales pinned to 1.0): inputs
    are exact e4m3, so vs the dequantized fp32 torch reference only
    accumulation-order noise remains. Measured round 2: 1666-1746 TF
    @4096^3 on random operands (profiles/gemm_fp8_mx.json)."""
    import torch

    torch.manual_seed(9)
    A = (torch.randn(512, 512, device="cuda") * 0.5).to(torch.float8_e4m3fn)
    Bt = (torch.randn(256, 512, device="cuda") * 0.5).to(torch.float8_e4m3fn)
    ref = A.float() @ Bt.float().T
    scale = ref.abs().max().item() + 1e-6
    for _ in range(3):  # race screen: nondeterminism would betray a pipeline bug
        C = ext.gemm_fp8_mx(A.view(torch.uint8), Bt.view(torch.uint8))
        err = (C - ref).abs().max().item() / scale
        assert err < 1e-3, f"fp8 MX rel err {err}"
    tflops = ext.gemm_fp8_mx_tflops(0, 4096, 8)
    print(f"fp8 MX GEMM (8-phase 256^2, K=128): {tflops:.0f} TFLOPS @4096^3")
    assert tflops > 1380, f"fp8 MX GEMM regressed: {tflops} TF (floor 1380 = 75% of measured 1849 post-swizzle-fix)"


def test_gemm_fp4_mx_numerics_and_throughput(ext):
    """MX-fp4 (e2m1, nibble-packed, K-step 256): every e2m1 value is a
    small multiple of 0.5, so fp32 accumulation is EXACT — the kernel must
    match the LUT-dequantized torch reference bit-for-bit. Measured round
    2: 3086-3213 TF @4096^3, ~4.0 PF @8192^3 (profiles/gemm_fp8_mx.json)."""
    import torch

    lut = torch.tensor(
        [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
         -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0], device="cuda")
    gen = torch.Generator(device="cuda").manual_seed(13)
    nA = torch.randint(0, 16, (512, 1024), generator=gen, device="cuda", dtype=torch.uint8)
    nB = torch.randint(0, 16, (256, 1024), generator=gen, device="cuda", dtype=torch.uint8)
    packA = (nA[:, 0::2] | (nA[:, 1::2] << 4)).contiguous()
    packB = (nB[:, 0::2] | (nB[:, 1::2] << 4)).contiguous()
    ref = lut[nA.long()] @ lut[nB.long()].T
    for _ in range(3):  # race screen
        C = ext.gemm_fp4_mx(packA, packB, 1024)
        torch.cuda.synchronize()
        assert torch.equal(C, ref), "fp4 MX result not exact"
    tflops = ext.gemm_fp4_mx_tflops(0, 4096, 8)
    print(f"fp4 MX GEMM: {tflops:.0f} TFLOPS @4096^3")
    assert tflops > 2300, f"fp4 MX GEMM regressed: {tflops} TF (floor 2300 = 75% of measured 3086)"


def test_tenant_gpu_workload_end_to_end(tmp_path, run):
    """Capstone e2e: a replicaSet whose WORKLOAD is a real torch compute
    job on its allocated GPU — the control plane must deliver a container
    in which CUDA-on-ROCm works and the matmul result is correct."""
    require_gpu()
    import sys
    import time as _time

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from helpers import make_daemon

    from gpu_docker_api_amd.models import ContainerRun

    marker = str(tmp_path / "result.txt")
    workload = (
        "import torch; assert torch.cuda.is_available();"
        "a = torch.ones(512, 512, device='cuda');"
        "v = (a @ a).sum().item();"
        f"open({marker!r}, 'w').write(str(v))"
    )

    async def main():
        d = await make_daemon(tmp_path, runtime="proc", inventory="amdsmi", copy_engine="auto")
        await d.replicaset.run_gpu_container(
            ContainerRun(
                image_name="synthetic:workload",
                replica_set_name="job",
                gpu_count=1,
                cpu_count=2,
                cmd=["python", "-c", workload],
            )
        )
        deadline = _time.time() + 120  # first torch import on a fresh box is slow
        while _time.time() < deadline and not os.path.exists(marker):
            await asyncio.sleep(0.5)
        assert os.path.exists(marker), "workload never wrote its result"
        assert float(open(marker).read()) == 512.0 * 512 * 512
        await d.replicaset.delete_container("job")
        await d.stop()

    import asyncio

    run(main())


def test_numa_local_cpuset_on_hardware(tmp_path, run):
    """A 1-GPU container's cpuset must come from the GPU's NUMA node
    (sysfs truth) when the host exposes multiple nodes."""
    require_gpu()
    import sys

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from helpers import make_daemon

    from gpu_docker_api_amd.models import ContainerRun
    from gpu_docker_api_amd.parallel.numa import cpu_node_map

    async def main():
        node_of = cpu_node_map()
        d = await make_daemon(tmp_path, runtime="proc", inventory="amdsmi")
        gpu_nodes = {g.numa_node for g in d.gpu.gpus if g.numa_node >= 0}
        if len(set(node_of.values())) < 2 or not gpu_nodes:
            pytest.skip("host exposes no multi-node NUMA topology")
        await d.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="nm", gpu_count=1,
                         cpu_count=4)
        )
        st = await d.runtime.inspect("nm-1")
        cpus = [int(c) for c in st.cpuset_cpus.split(",")]
        alloc_nodes = {node_of.get(c) for c in cpus}
        spec_gpu_nodes = d.replicaset._gpu_numa_nodes(st.gpu_uuids)
        assert alloc_nodes == set(spec_gpu_nodes), (cpus, alloc_nodes, spec_gpu_nodes)
        await d.replicaset.delete_container("nm")
        await d.stop()

    run(main())
