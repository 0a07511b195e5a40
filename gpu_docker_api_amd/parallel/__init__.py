from .inventory import GpuInfo, GpuInventory, AmdSmiInventory, MockInventory, make_inventory
from .topology import Topology
from .gpu import GpuScheduler
from .cpu import CpuScheduler
from .ports import PortScheduler

__all__ = [
    "GpuInfo",
    "GpuInventory",
    "AmdSmiInventory",
    "MockInventory",
    "make_inventory",
    "Topology",
    "GpuScheduler",
    "CpuScheduler",
    "PortScheduler",
]
