"""Host NUMA topology: cpu id -> node map.

MI355X nodes hang GPUs off specific sockets; a container's cpuset should
come from the same node(s) as its GPUs or every host<->device transfer
crosses the socket interconnect. The reference allocates CPUs by bare
index with no locality notion (cpuscheduler.go:77-116).
"""
from __future__ import annotations

import glob
import os
import re
from typing import Dict, List


def _parse_cpulist(text: str) -> List[int]:
    """'0-3,8,10-11' -> [0,1,2,3,8,10,11]"""
    out: List[int] = []
    for part in text.strip().split(","):
        part = part.strip()
        if not part:
            continue
        if "-" in part:
            a, b = part.split("-", 1)
            out.extend(range(int(a), int(b) + 1))
        else:
            out.append(int(part))
    return out


def cpu_node_map() -> Dict[int, int]:
    """cpu id -> NUMA node from sysfs; {} when the host exposes none."""
    out: Dict[int, int] = {}
    for node_dir in glob.glob("/sys/devices/system/node/node[0-9]*"):
        m = re.search(r"node(\d+)$", node_dir)
        if not m:
            continue
        node = int(m.group(1))
        try:
            with open(os.path.join(node_dir, "cpulist")) as f:
                for cpu in _parse_cpulist(f.read()):
                    out[cpu] = node
        except (OSError, ValueError):
            continue
    return out
