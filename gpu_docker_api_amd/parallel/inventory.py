"""GPU inventory backends.

The reference enumerates GPUs by forking ``nvidia-smi --query-gpu=index,uuid``
once at startup (/root/reference/internal/schedulers/gpuscheduler.go:20,167-179)
and knows nothing else about them. Here the inventory is a first-class
interface carrying what an MI355X control plane actually needs: UUID, PCIe
BDF, DRM render/card nodes (for device injection), HBM total/used bytes
(288 GB per MI355X), and per-pair link info for topology-aware placement.

Backends (config-selected — the reference needs a separate compiled binary
per backend via Go build tags):

* :class:`AmdSmiInventory` — the amdsmi Python library (libamd_smi), with an
  ``amd-smi`` CLI JSON fallback.
* :class:`MockInventory` — 8 fake MI355X with 288 GB HBM and a full 7-link
  xGMI mesh; the analog of the reference's mock flavor
  (gpuscheduler_mock.go:164-184) and what CPU-only tests run against.
"""
from __future__ import annotations

import glob
import json
import logging
import os
import subprocess
from dataclasses import dataclass
from typing import Dict, List, Optional

log = logging.getLogger(__name__)

MI355X_HBM_BYTES = 288 * 1024**3
XGMI_LINK_GBPS = 153.0  # per-link peer bandwidth, 7 links/GPU on an 8-GPU node
PCIE_GBPS = 32.0


@dataclass
class GpuInfo:
    index: int
    uuid: str
    bdf: str = ""
    name: str = "AMD Instinct MI355X"
    vram_total: int = MI355X_HBM_BYTES
    vram_used: int = 0
    numa_node: int = -1     # host NUMA node of the GPU (-1 = unknown)
    render_node: str = ""   # /dev/dri/renderD<N>
    card_node: str = ""     # /dev/dri/card<N>

    def to_dict(self) -> dict:
        return {
            "index": self.index,
            "uuid": self.uuid,
            "bdf": self.bdf,
            "name": self.name,
            "vramTotal": self.vram_total,
            "vramUsed": self.vram_used,
            "numaNode": self.numa_node,
            "renderNode": self.render_node,
            "cardNode": self.card_node,
        }


class GpuInventory:
    """Interface: enumerate GPUs + per-pair link bandwidth matrix."""

    def enumerate(self) -> List[GpuInfo]:
        raise NotImplementedError

    def refresh_usage(self) -> List[GpuInfo]:
        """Re-read volatile fields (HBM used). Default: re-enumerate."""
        return self.enumerate()

    def link_matrix(self) -> List[List[float]]:
        """Symmetric GB/s matrix; diagonal 0. Static topology estimate —
        measured numbers from the HIP probe overlay this (topology.py)."""
        raise NotImplementedError


def _bdf_numa_node(bdf: str) -> int:
    """Host NUMA node of a PCIe device (-1 when sysfs doesn't say)."""
    try:
        with open(f"/sys/bus/pci/devices/{bdf}/numa_node") as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return -1


def _bdf_to_drm_nodes(bdf: str) -> tuple[str, str]:
    """Resolve a PCIe BDF like '0000:05:00.0' to (/dev/dri/renderD*, /dev/dri/card*)
    by walking /sys/class/drm symlinks."""
    render, card = "", ""
    for node in glob.glob("/sys/class/drm/renderD*") + glob.glob("/sys/class/drm/card*"):
        try:
            dev = os.path.realpath(os.path.join(node, "device"))
        except OSError:
            continue
        if os.path.basename(dev).lower() == bdf.lower():
            name = os.path.basename(node)
            path = f"/dev/{'dri/' + name}"
            if name.startswith("renderD"):
                render = path
            else:
                card = path
    return render, card


class AmdSmiInventory(GpuInventory):
    """Real enumeration via the amdsmi library (preferred) or amd-smi CLI."""

    def __init__(self) -> None:
        self._lib = None
        self._handles: list = []
        try:
            import amdsmi  # type: ignore

            amdsmi.amdsmi_init()
            self._lib = amdsmi
            self._handles = amdsmi.amdsmi_get_processor_handles()
        except Exception as exc:  # library absent or no /dev/kfd
            log.info("amdsmi library unavailable (%s); will try amd-smi CLI", exc)

    def enumerate(self) -> List[GpuInfo]:
        if self._lib is not None and self._handles:
            return self._enumerate_lib()
        return self._enumerate_cli()

    def _enumerate_lib(self) -> List[GpuInfo]:
        smi = self._lib
        out: List[GpuInfo] = []
        for i, h in enumerate(self._handles):
            uuid = bdf = name = ""
            total, used = MI355X_HBM_BYTES, 0
            try:
                uuid = str(smi.amdsmi_get_gpu_device_uuid(h))
            except Exception:
                uuid = f"GPU-{i}"
            try:
                bdf = str(smi.amdsmi_get_gpu_device_bdf(h))
            except Exception:
                pass
            try:
                asic = smi.amdsmi_get_gpu_asic_info(h)
                name = str(asic.get("market_name") or asic.get("asic_serial") or "")
            except Exception:
                pass
            try:
                vu = smi.amdsmi_get_gpu_vram_usage(h)
                # amdsmi reports MB
                total = int(vu.get("vram_total", 0)) * 1024**2 or MI355X_HBM_BYTES
                used = int(vu.get("vram_used", 0)) * 1024**2
            except Exception:
                pass
            render, card = _bdf_to_drm_nodes(bdf) if bdf else ("", "")
            numa = _bdf_numa_node(bdf) if bdf else -1
            out.append(
                GpuInfo(
                    index=i,
                    uuid=uuid,
                    bdf=bdf,
                    name=name or "AMD Instinct MI355X",
                    vram_total=total,
                    vram_used=used,
                    numa_node=numa,
                    render_node=render,
                    card_node=card,
                )
            )
        return out

    def _enumerate_cli(self) -> List[GpuInfo]:
        """amd-smi CLI fallback. Wire formats (captured from a real MI355X,
        tests/fixtures/amdsmi_*.json):
          list:   [{"gpu": 0, "bdf": "0000:23:00.0", "uuid": "...", ...}]
          static: {"gpu_data": [{"gpu": 0, "asic": {"market_name": ...},
                   "vram": {"size": {"value": 294896, "unit": "MB"}}, ...}]}
        """
        try:
            raw = subprocess.run(
                ["amd-smi", "list", "--json"],
                capture_output=True,
                text=True,
                timeout=30,
                check=True,
            ).stdout
            data = json.loads(raw)
        except Exception as exc:
            raise RuntimeError(f"GPU enumeration failed (amdsmi lib and CLI): {exc}") from exc
        static_by_gpu = {}
        try:
            raw = subprocess.run(
                ["amd-smi", "static", "--json"],
                capture_output=True,
                text=True,
                timeout=60,
                check=True,
            ).stdout
            for g in json.loads(raw).get("gpu_data", []) or []:
                static_by_gpu[int(g.get("gpu", -1))] = g
        except Exception:
            pass  # enumeration works without static enrichment
        return self.parse_cli_output(data, static_by_gpu)

    @staticmethod
    def parse_cli_output(
        data, static_by_gpu: Optional[dict] = None
    ) -> List[GpuInfo]:
        static_by_gpu = static_by_gpu or {}
        out: List[GpuInfo] = []
        items = data if isinstance(data, list) else data.get("gpu", []) or []
        for i, item in enumerate(items):
            idx = int(item.get("gpu", i))
            bdf = str(item.get("bdf", ""))
            render, card = _bdf_to_drm_nodes(bdf) if bdf else ("", "")
            numa = _bdf_numa_node(bdf) if bdf else -1
            name = "AMD Instinct MI355X"
            total = MI355X_HBM_BYTES
            st = static_by_gpu.get(idx) or {}
            asic = st.get("asic") or {}
            if asic.get("market_name"):
                name = str(asic["market_name"])
            vram = (st.get("vram") or {}).get("size") or {}
            if isinstance(vram, dict) and vram.get("value"):
                unit = str(vram.get("unit", "MB")).upper()
                factor = {"KB": 1024, "MB": 1024**2, "GB": 1024**3}.get(unit, 1024**2)
                total = int(vram["value"]) * factor
            out.append(
                GpuInfo(
                    index=idx,
                    uuid=str(item.get("uuid", f"GPU-{i}")),
                    bdf=bdf,
                    name=name,
                    vram_total=total,
                    numa_node=numa,
                    render_node=render,
                    card_node=card,
                )
            )
        return out

    def link_matrix(self) -> List[List[float]]:
        n = len(self._handles)
        if self._lib is None or n == 0:
            gpus = self.enumerate()
            n = len(gpus)
            # no topology info from CLI path: assume full xGMI mesh on one node
            return [
                [0.0 if i == j else XGMI_LINK_GBPS for j in range(n)] for i in range(n)
            ]
        smi = self._lib
        mat = [[0.0] * n for _ in range(n)]
        for i in range(n):
            for j in range(n):
                if i == j:
                    continue
                gbps = PCIE_GBPS
                try:
                    lt = smi.amdsmi_topo_get_link_type(self._handles[i], self._handles[j])
                    type_val = str(lt.get("type", "")) if isinstance(lt, dict) else str(lt)
                    if "XGMI" in type_val.upper():
                        gbps = XGMI_LINK_GBPS
                except Exception:
                    pass
                mat[i][j] = gbps
        return mat

    def close(self) -> None:
        if self._lib is not None:
            try:
                self._lib.amdsmi_shut_down()
            except Exception:
                pass


class MockInventory(GpuInventory):
    """8 fake MI355X, full xGMI mesh — config-selected fake backend."""

    def __init__(self, count: int = 8) -> None:
        self.count = count
        self._used: Dict[int, int] = {}

    def enumerate(self) -> List[GpuInfo]:
        return [
            GpuInfo(
                index=i,
                uuid=f"MockMI355X-{i}",
                bdf=f"0000:{0x10 + i:02x}:00.0",
                vram_used=self._used.get(i, 0),
                # synthetic NUMA topology (two sockets, 4 GPUs each) so
                # NUMA-aware cpuset placement is testable on CPU boxes
                numa_node=i // 4,
                # synthetic DRM nodes (renderD128+i mirrors amdgpu's
                # numbering) so the docker device-injection path is
                # exercised by the mock backend too
                render_node=f"/dev/dri/renderD{128 + i}",
                card_node=f"/dev/dri/card{i}",
            )
            for i in range(self.count)
        ]

    def link_matrix(self) -> List[List[float]]:
        return [
            [0.0 if i == j else XGMI_LINK_GBPS for j in range(self.count)]
            for i in range(self.count)
        ]


def make_inventory(kind: str = "auto", mock_count: int = 8) -> GpuInventory:
    """auto: amdsmi if a GPU is visible, else mock (so CPU-only boxes work)."""
    if kind == "mock":
        return MockInventory(mock_count)
    if kind == "amdsmi":
        return AmdSmiInventory()
    # auto
    if os.path.exists("/dev/kfd"):
        try:
            inv = AmdSmiInventory()
            if inv.enumerate():
                return inv
        except Exception as exc:
            log.warning("amdsmi enumeration failed (%s); falling back to mock", exc)
    return MockInventory(mock_count)
