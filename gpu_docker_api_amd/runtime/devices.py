"""ROCm device injection.

The reference attaches GPUs with ``DeviceRequests: [{Driver:"cdi",
DeviceIDs:[...]}]`` + ``HostConfig.Runtime = "nvidia"``
(/root/reference/internal/services/replicaset_nomock.go:128-140) — i.e. it
delegates everything to the NVIDIA container toolkit. ROCm needs no runtime
shim: a container sees GPU N iff it has ``/dev/kfd`` (compute) and that GPU's
``/dev/dri/renderD*`` node (and is in the video/render groups). So injection
here is plain device cgroup entries on the stock ``runc`` runtime — no dual
paths, no CDI requirement (CDI names like ``amd.com/gpu=N`` can be layered on
by callers that want them).
"""
from __future__ import annotations

import grp
from typing import Dict, List

from .base import GpuResolver

KFD = "/dev/kfd"


def docker_devices_for(uuids: List[str], resolver: GpuResolver) -> List[Dict[str, str]]:
    """HostConfig.Devices entries for the docker driver."""
    if not uuids:
        return []
    devices = [
        {"PathOnHost": KFD, "PathInContainer": KFD, "CgroupPermissions": "rwm"}
    ]
    for u in uuids:
        info = resolver(u)
        if info is None:
            continue
        for node in (info.render_node, info.card_node):
            if node:
                devices.append(
                    {"PathOnHost": node, "PathInContainer": node, "CgroupPermissions": "rwm"}
                )
    return devices


def docker_group_add() -> List[str]:
    """video/render groups (by gid when resolvable, else by name)."""
    out = []
    for name in ("video", "render"):
        try:
            out.append(str(grp.getgrnam(name).gr_gid))
        except KeyError:
            out.append(name)
    return out


def visible_device_env(uuids: List[str], resolver: GpuResolver) -> Dict[str, str]:
    """GPU isolation env for the proc runtime: ROCR_VISIBLE_DEVICES restricts
    the ROCm runtime to the allocated GPUs (by index; UUIDs also accepted by
    ROCm but indices survive amdsmi/HIP enumeration-order differences)."""
    idx: List[str] = []
    for u in uuids:
        info = resolver(u)
        idx.append(str(info.index) if info is not None else u)
    if not uuids:
        # explicit empty set = no GPUs visible (a 0-GPU container on a GPU box)
        return {"ROCR_VISIBLE_DEVICES": "", "HIP_VISIBLE_DEVICES": ""}
    csv = ",".join(idx)
    return {"ROCR_VISIBLE_DEVICES": csv, "HIP_VISIBLE_DEVICES": csv}
