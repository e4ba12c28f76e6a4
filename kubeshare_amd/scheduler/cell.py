"""Cell tree — the scheduler's model of the physical cluster.

A cell is a subtree of the hardware hierarchy: leaf = one GPU (level 1),
node-level cell = one host's xGMI clique, higher levels = multi-node
groupings (reference Cell struct cell.go:131-153, constructor
cell.go:214-286). State tracked per cell: fractional `available` GPU,
`free_memory`, `healthy`, and for leaves the physical `uuid`.

Reservation/reclaim walk leaf->root adjusting availability (reference
pod.go:479-526); node health toggles subtree `healthy` bits and bubbles
memory to parents (node.go:216-285).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from .topology import TopologyConfig, build_cell_elements


@dataclass
class Cell:
    cell_type: str
    id: str
    level: int
    priority: int
    node_name: str = ""        # host this subtree lives on ("" above nodes)
    parent: Optional["Cell"] = None
    children: list = field(default_factory=list)
    leaf_cell_number: int = 1
    # dynamic state
    healthy: bool = False
    available: float = 0.0     # fractional GPUs available in this subtree
    available_whole: int = 0   # whole free leaves in this subtree
    free_memory: int = 0
    full_memory: int = 0
    uuid: str = ""             # leaf only
    # leaf only: peer-uuid -> xGMI link count (None = unknown topology,
    # assume the MI355X 7-link clique; missing peer = link down/absent)
    xgmi_peers: Optional[dict] = None

    @property
    def is_leaf(self) -> bool:
        return self.level == 1

    def leaves(self):
        if self.is_leaf:
            yield self
        else:
            for c in self.children:
                yield from c.leaves()

    def __repr__(self):
        return (f"Cell({self.id} {self.cell_type} avail={self.available} "
                f"mem={self.free_memory} healthy={self.healthy})")


class CellTree:
    """All cells of the cluster, indexed by leaf GPU model and by UUID."""

    def __init__(self, cfg: TopologyConfig):
        self.elements, self.gpu_priority = build_cell_elements(cfg)
        # model -> list of root cells whose leaf type is that model
        self.roots_by_model: dict[str, list[Cell]] = {}
        self.leaf_by_uuid: dict[str, Cell] = {}
        self.leaf_by_id: dict[str, Cell] = {}
        self.node_cells: dict[str, list[Cell]] = {}  # node -> node-level cells
        for spec in cfg.cells:
            root = self._build(spec, parent=None)
            self.roots_by_model.setdefault(
                self.elements[root.cell_type].leaf_cell_type, []).append(root)
            for leaf in root.leaves():
                self.leaf_by_id[leaf.id] = leaf
        # models sorted by priority desc (reference sortGPUPriority
        # cell.go:57-72) — scheduling considers faster models first
        self.models_by_priority = sorted(
            self.gpu_priority, key=lambda m: -self.gpu_priority[m])

    def _build(self, spec, parent: Optional[Cell], idx: int = 0,
               node_name: str = "") -> Cell:
        el = self.elements[spec.cell_type] if spec.cell_type else None
        if el is None:
            raise ValueError(f"unknown cell type {spec.cell_type!r}")
        cell_id = spec.cell_id or (f"{parent.id}/{idx}" if parent else
                                   f"cell{idx}")
        if el.is_node and not node_name:
            node_name = cell_id  # node-level instance id IS the host name
        cell = Cell(cell_type=spec.cell_type, id=cell_id, level=el.level,
                    priority=el.priority, node_name=node_name, parent=parent,
                    leaf_cell_number=el.leaf_cell_number)
        if el.is_node:
            self.node_cells.setdefault(node_name, []).append(cell)
        # children: explicit specs or inferred full fan-out
        # (ID inference "parent/i", reference config.go:77-120)
        if el.child_cell_type:
            child_specs = spec.children
            if not child_specs:
                from .topology import CellSpec
                child_specs = [CellSpec(cell_type=el.child_cell_type)
                               for _ in range(el.child_cell_number)]
            for i, cs in enumerate(child_specs):
                if not cs.cell_type:
                    cs.cell_type = el.child_cell_type
                cell.children.append(
                    self._build(cs, cell, i, node_name))
        return cell

    # ------------------------------------------------------------ inventory
    def assign_node_inventory(self, node: str, gpus_by_model: dict,
                              healthy: bool = True):
        """Bind physical GPUs (uuid/memory) to this node's leaf cells and
        mark the subtree healthy (reference setCellStatus node.go:127-197,
        passMemoryToParent node.go:257-285). `gpus_by_model`:
        model -> list[{uuid, memory, index?}]."""
        for node_cell in self.node_cells.get(node, []):
            model = self.elements[node_cell.cell_type].leaf_cell_type
            gpus = list(gpus_by_model.get(model, []))
            leaves = [c for c in node_cell.leaves()]
            # node-local GPU index -> uuid, to translate the inventory's
            # index-keyed xGMI adjacency into uuid-keyed leaf peers
            idx2uuid = {g.get("index", i): g["uuid"]
                        for i, g in enumerate(gpus)}

            def fill(leaf, gpu, first_fill):
                leaf.uuid = gpu["uuid"]
                leaf.full_memory = int(gpu["memory"])
                links = gpu.get("xgmi_links")
                if links is not None:
                    leaf.xgmi_peers = {
                        idx2uuid[j]: w for j, w in links.items()
                        if j in idx2uuid and w > 0}
                if first_fill:
                    leaf.free_memory = leaf.full_memory
                    leaf.available = 1.0
                    leaf.available_whole = 1
                self.leaf_by_uuid[leaf.uuid] = leaf

            # UUID-STABLE assignment: a re-register must keep each
            # surviving GPU on the leaf that carries its reservations —
            # positional assignment would re-bind every leaf when an
            # earlier-indexed GPU drops out of the inventory.
            by_uuid = {c.uuid: c for c in leaves if c.uuid}
            new_gpus = []
            for gpu in gpus:
                leaf = by_uuid.get(gpu["uuid"])
                if leaf is None:
                    new_gpus.append(gpu)
                else:
                    fill(leaf, gpu, first_fill=False)
            empty = [c for c in leaves if not c.uuid]
            for leaf, gpu in zip(empty, new_gpus):
                fill(leaf, gpu, first_fill=True)
            self.set_subtree_health(node_cell, healthy)
            # per-GPU failure: a leaf whose GPU no longer appears in
            # the inventory (device dropped off amdsmi) must not stay
            # schedulable on a stale UUID (the reference only had
            # node-level health, node.go:95-254)
            present = {g["uuid"] for g in gpus}
            missing = [c for c in leaves if c.uuid and
                       c.uuid not in present]
            for leaf in missing:
                leaf.healthy = False
            if missing:
                p = node_cell
                while p is not None:
                    if p.children:
                        p.healthy = any(ch.healthy for ch in p.children)
                    p = p.parent
            self._recompute_up(node_cell)

    def set_subtree_health(self, cell: Cell, healthy: bool):
        cell.healthy = healthy
        for c in cell.children:
            self.set_subtree_health(c, healthy)
        # parents are healthy iff any child is
        p = cell.parent
        while p is not None:
            p.healthy = any(ch.healthy for ch in p.children)
            p = p.parent

    def set_node_health(self, node: str, healthy: bool):
        for node_cell in self.node_cells.get(node, []):
            self.set_subtree_health(node_cell, healthy)

    # ---------------------------------------------------------- allocation
    def reserve(self, leaf: Cell, request: float, memory: int):
        """Charge a reservation at `leaf` and propagate to ancestors
        (reference reserveResource pod.go:479-501)."""
        self._apply(leaf, -min(request, 1.0), -memory)

    def reclaim(self, leaf: Cell, request: float, memory: int):
        """Undo a reservation (reference reclaimResource pod.go:504-526)."""
        self._apply(leaf, min(request, 1.0), memory)

    def _apply(self, leaf: Cell, d_avail: float, d_mem: int):
        if not leaf.is_leaf:
            raise ValueError("reserve/reclaim operate on leaf cells")
        whole_before = 1 if leaf.available >= 1.0 else 0
        avail = min(1.0, max(0.0, leaf.available + d_avail))
        # snap float drift: 1.0 - 0.3 - 0.1 + 0.1 + 0.3 ends at
        # 0.9999999999999999, which would permanently hide the leaf
        # from whole-GPU pods (property-tested in test_scheduler.py)
        if abs(avail - 1.0) < 1e-9:
            avail = 1.0
        elif avail < 1e-9:
            avail = 0.0
        leaf.available = avail
        leaf.free_memory = max(0, min(leaf.full_memory,
                                      leaf.free_memory + d_mem))
        whole_after = 1 if leaf.available >= 1.0 else 0
        leaf.available_whole = whole_after
        d_whole = whole_after - whole_before
        p = leaf.parent
        while p is not None:
            p.available += d_avail
            p.free_memory += d_mem
            p.available_whole += d_whole
            p = p.parent

    def _recompute_up(self, cell: Cell):
        """Recompute aggregates bottom-up after inventory assignment."""
        if not cell.is_leaf:
            for c in cell.children:
                self._recompute_up(c)
            cell.available = sum(c.available for c in cell.children)
            cell.available_whole = sum(c.available_whole
                                       for c in cell.children)
            cell.free_memory = sum(c.free_memory for c in cell.children)
            cell.full_memory = sum(c.full_memory for c in cell.children)
        p = cell.parent
        while p is not None:
            p.available = sum(c.available for c in p.children)
            p.available_whole = sum(c.available_whole for c in p.children)
            p.free_memory = sum(c.free_memory for c in p.children)
            p.full_memory = sum(c.full_memory for c in p.children)
            p = p.parent

    # -------------------------------------------------------------- lookup
    def leaves_on_node(self, node: str, model: str = "") -> list[Cell]:
        """Healthy leaf cells on `node`, optionally pinned to a GPU model
        (reference getModelLeafCellbyNode/getAllLeafCellbyNode
        score.go:230-294). Order: model priority desc, then cell id."""
        out = []
        models = [model] if model else self.models_by_priority
        for m in models:
            for node_cell in self.node_cells.get(node, []):
                if self.elements[node_cell.cell_type].leaf_cell_type != m:
                    continue
                out.extend(c for c in node_cell.leaves()
                           if c.healthy and c.uuid)
        return out

    def all_nodes(self) -> list[str]:
        return list(self.node_cells)
