"""Cluster topology config — the kubeshare-config.yaml contract.

Format kept verbatim from the reference (pkg/scheduler/config.go:15-35,
examples deploy/config/*.yaml; authoring guide doc/deploy.md:25-128):

    cellTypes:
      MI355X-NODE:
        childCellType: "AMD Instinct MI355X"
        childCellNumber: 8
        childCellPriority: 100
        isNodeLevel: true
      2-MI355X-NODE:
        childCellType: MI355X-NODE
        childCellNumber: 2
    cells:
    - cellType: 2-MI355X-NODE
      cellChildren:
      - cellId: node-a
      - cellId: node-b

Cell IDs are auto-inferred "parent/i" below the instance level
(reference config.go:77-120). A cell type not present in cellTypes is a
LEAF (= one physical GPU model); node-level cells are the per-node xGMI
clique (8 GPUs, 7 p2p links each on MI355X).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import yaml


@dataclass
class CellTypeSpec:
    child_cell_type: str
    child_cell_number: int
    child_cell_priority: int = 0
    is_node_level: bool = False


@dataclass
class CellSpec:
    cell_type: str = ""
    cell_id: str = ""
    children: list = field(default_factory=list)


@dataclass
class TopologyConfig:
    cell_types: dict  # name -> CellTypeSpec
    cells: list       # list[CellSpec]

    @classmethod
    def from_yaml(cls, text: str) -> "TopologyConfig":
        raw = yaml.safe_load(text) or {}
        types = {}
        for name, spec in (raw.get("cellTypes") or {}).items():
            types[name] = CellTypeSpec(
                child_cell_type=spec.get("childCellType", ""),
                child_cell_number=int(spec.get("childCellNumber", 0)),
                child_cell_priority=int(spec.get("childCellPriority", 0)),
                is_node_level=bool(spec.get("isNodeLevel", False)),
            )

        def parse_cell(d) -> CellSpec:
            return CellSpec(
                cell_type=d.get("cellType", ""),
                cell_id=d.get("cellId", ""),
                children=[parse_cell(c) for c in d.get("cellChildren", [])],
            )

        cells = [parse_cell(c) for c in (raw.get("cells") or [])]
        return cls(cell_types=types, cells=cells)

    @classmethod
    def from_file(cls, path: str) -> "TopologyConfig":
        with open(path) as f:
            return cls.from_yaml(f.read())

    @classmethod
    def single_node(cls, node: str, gpus: int = 8,
                    model: str = "AMD Instinct MI355X",
                    priority: int = 100) -> "TopologyConfig":
        """Convenience: one MI355X node (the common case)."""
        return cls(
            cell_types={
                "MI355X-NODE": CellTypeSpec(model, gpus, priority, True)},
            cells=[CellSpec(cell_type="MI355X-NODE", cell_id=node)],
        )

    @classmethod
    def nodes(cls, node_names, gpus: int = 8,
              model: str = "AMD Instinct MI355X",
              priority: int = 100) -> "TopologyConfig":
        """Convenience: a flat multi-node MI355X cluster (one node-level
        cell per host; cross-node locality uses the hierarchical cell-ID
        distance)."""
        return cls(
            cell_types={
                "MI355X-NODE": CellTypeSpec(model, gpus, priority, True)},
            cells=[CellSpec(cell_type="MI355X-NODE", cell_id=n)
                   for n in node_names],
        )


@dataclass
class CellElement:
    """Preprocessed per-type info (reference cell.go:34-129)."""
    cell_type: str
    level: int
    priority: int
    child_cell_type: str
    child_cell_number: int
    leaf_cell_type: str
    leaf_cell_number: int
    is_node: bool
    is_multi_node: bool


def build_cell_elements(cfg: TopologyConfig) -> tuple[dict, dict]:
    """Returns (elements by type, gpu priority by leaf model)."""
    elements: dict[str, CellElement] = {}
    gpu_priority: dict[str, int] = {}

    def add(cell_type: str, priority: int):
        if cell_type in elements:
            return
        spec: Optional[CellTypeSpec] = cfg.cell_types.get(cell_type)
        if spec is None:  # leaf = physical GPU model
            elements[cell_type] = CellElement(
                cell_type=cell_type, level=1, priority=priority,
                child_cell_type="", child_cell_number=0,
                leaf_cell_type=cell_type, leaf_cell_number=1,
                is_node=False, is_multi_node=False)
            gpu_priority[cell_type] = priority
            return
        add(spec.child_cell_type, spec.child_cell_priority)
        child = elements[spec.child_cell_type]
        elements[cell_type] = CellElement(
            cell_type=cell_type, level=child.level + 1,
            priority=child.priority,
            child_cell_type=child.cell_type,
            child_cell_number=spec.child_cell_number,
            leaf_cell_type=child.leaf_cell_type,
            leaf_cell_number=child.leaf_cell_number * spec.child_cell_number,
            is_node=spec.is_node_level,
            is_multi_node=child.is_node or child.is_multi_node)

    for t in cfg.cell_types:
        add(t, 0)
    return elements, gpu_priority
