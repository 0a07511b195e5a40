"""Resource routes (reference: internal/routers/resource.go:11-38), extended
with what an MI355X operator actually needs: per-GPU HBM free/used bytes and
the xGMI adjacency matrix (SURVEY.md §2.4 last row)."""
from __future__ import annotations

from fastapi import APIRouter, Request

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

    @r.post("/gpus/validate")
    async def gpus_validate(request: Request):
        """MI355X extension: burn-in the node's FREE GPUs (HBM bandwidth +
        dense bf16 MFMA GEMM) before trusting them with placements. Busy
        GPUs are skipped unless explicitly listed."""
        try:
            body = await request.json()
        except Exception:
            body = {}
        if not isinstance(body, dict):
            from .codes import Code
            from .response import error

            return error(Code.INVALID_PARAMS)
        requested = body.get("gpus")
        status = gpu.get_gpu_status()
        if requested:
            idx = [
                g.index
                for u in requested
                if (g := gpu.info_by_uuid(u)) is not None
            ]
        else:
            idx = [
                g.index for g in gpu.gpus if status.get(g.uuid, 1) == 0
            ]
        try:
            size = int(body.get("size", 4096))
            iters = int(body.get("iters", 5))
        except (TypeError, ValueError):
            from .codes import Code
            from .response import error

            return error(Code.INVALID_PARAMS)
        try:
            from ..ops import hipcore

            report = await hipcore.validate_gpus_async(
                idx, max(256, min(size, 16384)), max(1, min(iters, 50))
            )
        except Exception as exc:  # noqa: BLE001 — no GPU / extension absent
            from .codes import Code
            from .response import error

            return error(Code.SERVER_BUSY, detail=f"validation unavailable: {exc}")
        return success(report)

    @r.get("/cpus")
    async def cpus():
        return success(cpu.get_cpu_status())

    @r.get("/ports")
    async def port_status():
        return success(ports.get_port_status())

    return r
