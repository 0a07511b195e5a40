"""In-tree native build: hipcc for gfx950, no JIT cache.

Artifacts (git-ignored, but shipped by the repo snapshot to GPU boxes):

  gpu_docker_api_amd/ops/_hipcore.so   - torch extension (csrc/hipcore.hip)
  gpu_docker_api_amd/ops/_iocopy.so    - pybind11 module (csrc/iocopy.cpp)
  csrc/bin/rccl_smoke                  - standalone binary (csrc/rccl_smoke.hip)

Run: ``python -m gpu_docker_api_amd.ops.build`` (or __graft_entry__.build()).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
CSRC = os.path.join(ROOT, "csrc")
OPS = os.path.join(ROOT, "gpu_docker_api_amd", "ops")
BIN = os.path.join(CSRC, "bin")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _newer(out: str, *srcs: str) -> bool:
    if not os.path.exists(out):
        return False
    omt = os.path.getmtime(out)
    return all(os.path.getmtime(s) <= omt for s in srcs)


def _run(cmd: list[str]) -> None:
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build_hipcore(force: bool = False) -> str:
    src = os.path.join(CSRC, "hipcore.hip")
    includes = [
        os.path.join(CSRC, "gemm_bf16.hip"),
        os.path.join(CSRC, "gemm_bf16_8phase.hip"),
    ]
    out = os.path.join(OPS, "_hipcore.so")
    if not force and _newer(out, src, *[p for p in includes if os.path.exists(p)]):
        return out
    import torch
    from torch.utils import cpp_extension

    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    inc = cpp_extension.include_paths()
    lib = cpp_extension.library_paths()
    py_inc = sysconfig.get_paths()["include"]
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        src,
        "-o",
        out,
        *[f"-I{p}" for p in inc],
        f"-I{py_inc}",
        *[f"-L{p}" for p in lib],
        "-ltorch",
        "-ltorch_python",
        "-lc10",
        "-ltorch_hip",
        "-lc10_hip",
        "-lamdhip64",
        "-DTORCH_EXTENSION_NAME=_hipcore",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DUSE_ROCM",
    ]
    _run(cmd)
    return out


def build_iocopy(force: bool = False) -> str:
    src = os.path.join(CSRC, "iocopy.cpp")
    out = os.path.join(OPS, "_iocopy.so")
    if not force and _newer(out, src):
        return out
    import pybind11

    py_inc = sysconfig.get_paths()["include"]
    cmd = [
        "g++",
        "-O2",
        "-std=c++17",
        "-fPIC",
        "-shared",
        src,
        "-o",
        out,
        f"-I{pybind11.get_include()}",
        f"-I{py_inc}",
    ]
    _run(cmd)
    return out


def build_rccl_smoke(force: bool = False) -> str:
    src = os.path.join(CSRC, "rccl_smoke.hip")
    out = os.path.join(BIN, "rccl_smoke")
    if not force and _newer(out, src):
        return out
    os.makedirs(BIN, exist_ok=True)
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O2",
        "-std=c++17",
        src,
        "-o",
        out,
        "-I/opt/rocm/include",
        "-L/opt/rocm/lib",
        "-lrccl",
    ]
    _run(cmd)
    return out


def build_all(force: bool = False) -> dict:
    return {
        "iocopy": build_iocopy(force),
        "hipcore": build_hipcore(force),
        "rccl_smoke": build_rccl_smoke(force),
    }


if __name__ == "__main__":
    force = "--force" in sys.argv
    for name, path in build_all(force).items():
        print(f"built {name}: {path}")
