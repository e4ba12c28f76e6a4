from .harness import FakeCluster, FakePod
from .inventory import AmdSmiInventory, FakeInventory, GPUInfo, TorchInventory
from .plugin import KubeShareScheduler, Placement, QueuedPodInfo
from .topology import CellSpec, CellTypeSpec, TopologyConfig

__all__ = [
    "AmdSmiInventory", "CellSpec", "CellTypeSpec", "FakeCluster", "FakePod",
    "FakeInventory", "GPUInfo", "KubeShareScheduler", "Placement",
    "QueuedPodInfo", "TopologyConfig", "TorchInventory",
]
