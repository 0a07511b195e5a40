"""Resource routes (reference: internal/routers/resource.go:11-38), extended
with what an MI355X operator actually needs: per-GPU HBM free/used bytes and
the xGMI adjacency matrix (SURVEY.md §2.4 last row)."""
from __future__ import annotations

from fastapi import APIRouter

from ..parallel import CpuScheduler, GpuScheduler, PortScheduler
from .response import success


def make_router(gpu: GpuScheduler, cpu: CpuScheduler, ports: PortScheduler) -> APIRouter:
    r = APIRouter(prefix="/api/v1/resources")

    @r.get("/gpus")
    async def gpus():
        # reference shape: the uuid -> 0|1 map verbatim
        return success(gpu.get_gpu_status())

    @r.get("/gpus/detail")
    async def gpus_detail():
        # MI355X extension: HBM bytes + xGMI adjacency for job sizing
        return success(gpu.get_detail())

    @r.get("/cpus")
    async def cpus():
        return success(cpu.get_cpu_status())

    @r.get("/ports")
    async def port_status():
        return success(ports.get_port_status())

    return r
