from .harness import FakeCluster, FakePod
from .inventory import AmdSmiInventory, FakeInventory, GPUInfo, TorchInventory
from .kube import KubeDriver
from .kubeclient import K8sObj, RestCoreV1, RestCustomObjects
from .plugin import KubeShareScheduler, Placement, QueuedPodInfo
from .topology import CellSpec, CellTypeSpec, TopologyConfig

__all__ = [
    "AmdSmiInventory", "CellSpec", "CellTypeSpec", "FakeCluster", "FakePod",
    "FakeInventory", "GPUInfo", "K8sObj", "KubeDriver",
    "KubeShareScheduler", "Placement", "QueuedPodInfo", "RestCoreV1",
    "RestCustomObjects", "TopologyConfig", "TorchInventory",
]
