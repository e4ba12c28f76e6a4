"""All shipped YAML (deploy manifests, CRD, examples, topology) must be
well-formed, and key references must be self-consistent."""
import glob
import os

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _load_all(path):
    with open(path) as f:
        return [d for d in yaml.safe_load_all(f) if d is not None]


def test_all_yaml_parses():
    files = (glob.glob(os.path.join(REPO, "deploy", "**", "*.yaml"),
                       recursive=True)
             + glob.glob(os.path.join(REPO, "crd", "*.yaml"))
             + glob.glob(os.path.join(REPO, "examples", "*.yaml")))
    assert len(files) >= 10
    for path in files:
        docs = _load_all(path)
        assert docs, path


def test_examples_use_our_scheduler():
    from kubeshare_amd.utils import constants as C
    for path in glob.glob(os.path.join(REPO, "examples", "pod-*.yaml")):
        for doc in _load_all(path):
            assert doc["spec"]["schedulerName"] == C.SCHEDULER_NAME, path


def test_topology_example_loads_into_cell_tree():
    from kubeshare_amd.scheduler.cell import CellTree
    from kubeshare_amd.scheduler.topology import TopologyConfig
    cfg = TopologyConfig.from_file(os.path.join(
        REPO, "deploy", "config", "kubeshare-config-mi355x.yaml"))
    tree = CellTree(cfg)
    assert set(tree.node_cells) == {"mi355x-node-a", "mi355x-node-b"}
    assert sum(1 for c in tree.node_cells["mi355x-node-a"][0].leaves()) == 8


def test_crd_matches_sharepod_controller():
    import kubeshare_amd.sharepod as sp
    docs = _load_all(os.path.join(REPO, "crd", "sharepod.yaml"))
    crd = docs[0]
    assert crd["spec"]["group"] == sp.SharePodController.GROUP
    assert crd["spec"]["names"]["plural"] == sp.SharePodController.PLURAL
    served = [v["name"] for v in crd["spec"]["versions"] if v["served"]]
    assert sp.SharePodController.VERSION in served


def test_console_scripts_resolve():
    """Every [project.scripts] entry point must import and be callable
    (packaging bitrot guard)."""
    import importlib

    import tomli

    with open(os.path.join(REPO, "pyproject.toml"), "rb") as f:
        cfg = tomli.load(f)
    scripts = cfg["project"]["scripts"]
    assert len(scripts) >= 3
    for target in scripts.values():
        mod, fn = target.split(":")
        assert callable(getattr(importlib.import_module(mod), fn)), target
