"""GPU inventory providers.

The reference scheduler reads its own inventory through Prometheus
inside the Filter hot path (gpu.go:22-53, scheduler.go:335) — an
acknowledged flaw (README.md:141 "Modify the prometheus to etcd"). Here
inventory is a provider interface injected into the scheduler; the
Prometheus exporter (kubeshare_amd.collector) remains for observability
but is not scheduler-critical.

Providers:
  - FakeInventory: unit tests / kind clusters with no GPUs.
  - AmdSmiInventory: real MI355X nodes via the amdsmi python bindings —
    uuid, model, memory AND the xGMI link graph (link type/weight/hops),
    which feeds the topology-aware Score term.
  - TorchInventory: fallback via torch.cuda device properties.
"""
from __future__ import annotations

from dataclasses import dataclass, field

from ..utils import constants as C


@dataclass
class GPUInfo:
    uuid: str
    model: str
    memory: int
    index: int
    numa_node: int = -1
    # xGMI adjacency: peer index -> number of links (MI355X: 7 p2p links
    # per GPU, ~153 GB/s each; 0 = not directly connected)
    xgmi_links: dict = field(default_factory=dict)


class FakeInventory:
    """node -> list[GPUInfo]; fully-connected xGMI clique per node by
    default (the MI355X single-hop topology)."""

    def __init__(self, nodes: dict | None = None):
        self._nodes: dict[str, list[GPUInfo]] = {}
        for node, spec in (nodes or {}).items():
            self.add_node(node, **spec)

    def add_node(self, node: str, gpus: int = 8,
                 model: str = C.MI355X_MODEL,
                 memory: int = C.MI355X_HBM_BYTES,
                 down_links: list | None = None):
        """`down_links`: list of (i, j) GPU-index pairs whose direct
        xGMI link is absent/failed (removed from both directions) — lets
        tests model a degraded topology."""
        down = set()
        for i, j in down_links or []:
            down.add((i, j))
            down.add((j, i))
        infos = []
        for i in range(gpus):
            links = {j: 1 for j in range(gpus)
                     if j != i and (i, j) not in down}
            infos.append(GPUInfo(uuid=f"GPU-{node}-{i}", model=model,
                                 memory=memory, index=i, xgmi_links=links))
        self._nodes[node] = infos

    def gpus(self, node: str) -> list[GPUInfo]:
        return list(self._nodes.get(node, []))

    def by_model(self, node: str) -> dict:
        out: dict[str, list] = {}
        for g in self.gpus(node):
            out.setdefault(g.model, []).append(
                {"uuid": g.uuid, "memory": g.memory, "index": g.index,
                 "xgmi_links": dict(g.xgmi_links)})
        return out


def format_node_annotation(gpus: list) -> str:
    """Serialize local inventory (list[GPUInfo]) into the
    `kubeshare.amd/gpus` node annotation the KubeDriver consumes:
    "uuid,model,memory,index,links=j:k;..." — links are the peer GPU
    indices with a live direct xGMI link."""
    parts = []
    for g in gpus:
        entry = f"{g.uuid},{g.model},{g.memory},{g.index}"
        if g.xgmi_links:
            links = ":".join(str(j) for j, w in sorted(g.xgmi_links.items())
                             if w > 0)
            entry += f",links={links}"
        # no link info -> omit the field entirely: an EMPTY links= would
        # read back as "every link down" instead of "unknown topology,
        # assume the MI355X clique"
        parts.append(entry)
    return ";".join(parts)


class AmdSmiInventory:
    """Local-node inventory via amdsmi (the collector daemonset runs one
    per node; the scheduler consumes its export)."""

    def __init__(self):
        import amdsmi
        self._smi = amdsmi
        amdsmi.amdsmi_init()

    def local_gpus(self) -> list[GPUInfo]:
        smi = self._smi
        handles = smi.amdsmi_get_processor_handles()
        infos = []
        for i, h in enumerate(handles):
            try:
                asic = smi.amdsmi_get_gpu_asic_info(h)
                model = asic.get("market_name") or C.MI355X_MODEL
            except Exception:  # noqa: BLE001
                model = C.MI355X_MODEL
            try:
                uuid = smi.amdsmi_get_gpu_device_uuid(h)
            except Exception:  # noqa: BLE001
                uuid = f"GPU-{i}"
            try:
                mem = smi.amdsmi_get_gpu_memory_total(
                    h, smi.AmdSmiMemoryType.VRAM)
            except Exception:  # noqa: BLE001
                mem = C.MI355X_HBM_BYTES
            info = GPUInfo(uuid=uuid, model=model, memory=int(mem), index=i)
            infos.append(info)
        # xGMI link graph (feeds the Score locality term)
        for i, hi in enumerate(handles):
            for j, hj in enumerate(handles):
                if i == j:
                    continue
                try:
                    link = self._smi.amdsmi_topo_get_link_type(hi, hj)
                    # XGMI == direct p2p; weight/hops vary by SKU
                    if str(link.get("type", "")).upper().find("XGMI") >= 0 \
                            or link.get("hops", 99) <= 1:
                        infos[i].xgmi_links[j] = 1
                except Exception:  # noqa: BLE001
                    # assume the MI355X clique when topo query fails
                    infos[i].xgmi_links[j] = 1
        return infos


class TorchInventory:
    def local_gpus(self) -> list[GPUInfo]:
        import torch
        infos = []
        for i in range(torch.cuda.device_count()):
            p = torch.cuda.get_device_properties(i)
            uuid = getattr(p, "uuid", None)
            infos.append(GPUInfo(
                uuid=str(uuid) if uuid else f"GPU-{i}",
                model=C.MI355X_MODEL if "MI355" in p.name or
                      p.name == "AMD Radeon Graphics" else p.name,
                memory=p.total_memory, index=i,
                xgmi_links={j: 1 for j in range(torch.cuda.device_count())
                            if j != i}))
        return infos
