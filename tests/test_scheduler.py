"""Scheduler-core tests: cell tree, filter/score/reserve/permit, gangs,
restart resync — covering BASELINE.json configs #1, #3, #4, #5 on the
in-memory harness (the fake-cluster layer SURVEY.md §4 calls for)."""
import pytest

from kubeshare_amd.scheduler.harness import FakeCluster
from kubeshare_amd.scheduler.topology import TopologyConfig
from kubeshare_amd.scheduler.cell import CellTree
from kubeshare_amd.utils import constants as C


def shared(request, limit="1.0", **extra):
    labels = {C.POD_GPU_REQUEST: request, C.POD_GPU_LIMIT: limit}
    labels.update(extra)
    return labels


# ---------------------------------------------------------------- topology
def test_cell_tree_from_yaml():
    cfg = TopologyConfig.from_yaml("""
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
""")
    tree = CellTree(cfg)
    assert set(tree.node_cells) == {"node-a", "node-b"}
    leaves = list(tree.node_cells["node-a"][0].leaves())
    assert len(leaves) == 8
    # auto-inferred ids "parent/i" (reference config.go:77-120)
    assert leaves[0].id == "node-a/0" and leaves[7].id == "node-a/7"
    assert tree.gpu_priority["AMD Instinct MI355X"] == 100


def test_inventory_assignment_bubbles_memory():
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    tree = fc.scheduler.tree
    node_cell = tree.node_cells["node-a"][0]
    assert node_cell.free_memory == 2 * C.MI355X_HBM_BYTES
    assert node_cell.available == 2.0
    assert len(tree.leaf_by_uuid) == 2


# -------------------------------------------------- config #1: one 0.5 pod
def test_single_half_pod_schedules():
    fc = FakeCluster(nodes={"node-a": {"gpus": 1}})
    pod = fc.add_pod("default", "p1", shared("0.5"))
    fc.schedule_pending()
    assert pod.phase == "Bound"
    assert pod.annotations[C.POD_GPU_UUID] == "GPU-node-a-0"
    assert pod.annotations[C.POD_CELL_ID] == "node-a/0"
    # default gpu_mem = floor(0.5 * 288 GiB) (reference pod.go:419-421)
    assert int(pod.annotations[C.POD_GPU_MEMORY]) == C.MI355X_HBM_BYTES // 2
    port = int(pod.annotations[C.POD_MANAGER_PORT])
    assert C.POD_MANAGER_PORT_START <= port < \
        C.POD_MANAGER_PORT_START + C.POD_MANAGER_PORT_POOL
    # ROCm-native env injection
    assert pod.env[C.ENV_ROCR_VISIBLE_DEVICES] == "0"
    assert pod.env[C.ENV_LD_PRELOAD] == C.HOOK_SO_PATH
    assert pod.env[C.ENV_POD_NAME] == "default/p1"
    # tree charged
    leaf = fc.scheduler.tree.leaf_by_uuid["GPU-node-a-0"]
    assert leaf.available == pytest.approx(0.5)


def test_invalid_labels_rejected():
    fc = FakeCluster()
    pod = fc.add_pod("default", "bad",
                     shared("0.5", limit="0.3"))  # limit < request
    fc.schedule_pending()
    assert pod.phase == "Unschedulable"


def test_regular_pod_ignored():
    fc = FakeCluster()
    pod = fc.add_pod("default", "plain", {})
    fc.schedule_pending()
    assert pod.phase == "Regular"


# ---------------------------------------------- packing / scoring behavior
def test_opportunistic_pods_pack_same_gpu():
    """Two opportunistic 0.3 pods must land on the SAME GPU
    (defragmentation packing, reference score.go:42-68)."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 4}})
    p1 = fc.add_pod("default", "o1", shared("0.3"))
    fc.schedule_pending()
    p2 = fc.add_pod("default", "o2", shared("0.3"))
    fc.schedule_pending()
    assert p1.phase == p2.phase == "Bound"
    assert p1.annotations[C.POD_GPU_UUID] == p2.annotations[C.POD_GPU_UUID]


def test_guarantee_pods_spread():
    """Two Guarantee 0.5 pods (no group) spread to different free GPUs
    (reference score.go:85-112: usage term is negative)."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 4}})
    p1 = fc.add_pod("default", "g1",
                    shared("0.5", **{C.POD_PRIORITY: "100"}))
    fc.schedule_pending()
    p2 = fc.add_pod("default", "g2",
                    shared("0.5", **{C.POD_PRIORITY: "100"}))
    fc.schedule_pending()
    assert p1.annotations[C.POD_GPU_UUID] != p2.annotations[C.POD_GPU_UUID]


def test_memory_filter():
    fc = FakeCluster(nodes={"node-a": {"gpus": 1}})
    big = str(C.MI355X_HBM_BYTES + 1)
    pod = fc.add_pod("default", "m1",
                     shared("0.5", **{C.POD_GPU_MEMORY: big}))
    fc.schedule_pending()
    assert pod.phase == "Unschedulable"


def test_model_pinning_unknown_model():
    fc = FakeCluster()
    pod = fc.add_pod("default", "mp",
                     shared("0.5", **{C.POD_GPU_MODEL: "test"}))
    fc.schedule_pending()
    assert pod.phase == "Unschedulable"  # reference test/pod10.yaml


# ------------------------------------- config #3: mixed fractions, 8 GPUs
def test_bin_packing_mixed_fractions():
    fc = FakeCluster(nodes={"node-a": {"gpus": 8}})
    pods = []
    for i, req in enumerate(["0.25", "0.25", "0.5", "0.5", "1.0", "1.0"]):
        pods.append(fc.add_pod("default", f"mix{i}", shared(req)))
        fc.schedule_pending()
    assert all(p.phase == "Bound" for p in pods)
    total = sum(1.0 - c.available
                for c in fc.scheduler.tree.leaves_on_node("node-a"))
    assert total == pytest.approx(0.25 + 0.25 + 0.5 + 0.5 + 1.0 + 1.0)


def test_multi_gpu_pod():
    fc = FakeCluster(nodes={"node-a": {"gpus": 8}})
    pod = fc.add_pod("default", "mg",
                     {C.POD_GPU_REQUEST: "4.0", C.POD_GPU_LIMIT: "4.0"})
    fc.schedule_pending()
    assert pod.phase == "Bound"
    assert len(pod.annotations[C.POD_GPU_UUID].split(",")) == 4
    # whole-GPU pods bypass the isolation layer (reference pod.go:348-400)
    assert C.POD_MANAGER_PORT not in pod.annotations
    assert C.ENV_LD_PRELOAD not in pod.env
    assert len(pod.env[C.ENV_ROCR_VISIBLE_DEVICES].split(",")) == 4


def test_multi_gpu_insufficient():
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    pod = fc.add_pod("default", "mg8",
                     {C.POD_GPU_REQUEST: "8.0", C.POD_GPU_LIMIT: "8.0"})
    fc.schedule_pending()
    assert pod.phase == "Unschedulable"


# --------------------------------------------- config #4: gang of 4 (atomic)
def gang_labels(request, group, headcount, threshold="1.0", priority="100"):
    return shared(request, **{
        C.POD_GROUP_NAME: group,
        C.POD_GROUP_HEADCOUNT: str(headcount),
        C.POD_GROUP_THRESHOLD: threshold,
        C.POD_PRIORITY: priority,
    })


def test_gang_waits_then_binds_atomically():
    fc = FakeCluster(nodes={"node-a": {"gpus": 8}})
    pods = [fc.add_pod("default", f"g{i}", gang_labels("1.0", "team", 4))
            for i in range(4)]
    fc.schedule_pending()
    assert all(p.phase == "Bound" for p in pods), \
        [(p.name, p.phase) for p in pods]


def test_partial_gang_times_out_and_releases():
    fc = FakeCluster(nodes={"node-a": {"gpus": 8}})
    pods = [fc.add_pod("default", f"pg{i}", gang_labels("1.0", "part", 4))
            for i in range(2)]  # only 2 of 4 members exist
    fc.schedule_pending()
    # PreFilter rejects early: total group pods < minAvailable
    # (reference scheduler.go:315-321)
    assert all(p.phase == "Unschedulable" for p in pods)
    # resources must not stay reserved
    assert all(c.available == 1.0
               for c in fc.scheduler.tree.leaves_on_node("node-a"))


def test_gang_wait_timeout_reclaims():
    """3 of 4 members schedulable (one blocked by capacity): waiting
    members are rejected at timeout and resources reclaimed."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    pods = [fc.add_pod("default", f"t{i}", gang_labels("1.0", "tmo", 4))
            for i in range(4)]
    fc.schedule_pending()
    assert all(p.phase in ("Waiting", "Unschedulable", "Pending")
               for p in pods)
    fc.advance(100.0)
    assert all(p.phase != "Waiting" for p in pods)
    assert all(c.available == 1.0
               for c in fc.scheduler.tree.leaves_on_node("node-a"))


def test_gang_min_available_threshold():
    """headcount=4, threshold=0.5 -> minAvailable 2: two pods suffice."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 8}})
    pods = [fc.add_pod("default", f"h{i}",
                       gang_labels("1.0", "half", 4, threshold="0.5"))
            for i in range(2)]
    fc.schedule_pending()
    assert all(p.phase == "Bound" for p in pods)


def test_gang_locality_same_node():
    """Guarantee gang members prefer the node already hosting the group
    (locality term, reference score.go:85-112,164-227)."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 8}, "node-b": {"gpus": 8}})
    pods = [fc.add_pod("default", f"l{i}",
                       gang_labels("1.0", "loc", 4))
            for i in range(4)]
    fc.schedule_pending()
    nodes = {p.node for p in pods}
    assert all(p.phase == "Bound" for p in pods)
    assert len(nodes) == 1, f"gang split across {nodes}"


# ------------------- config #5: Guarantee + Opportunistic oversubscription
def test_oversubscription_priorities():
    """4 Guarantee 0.5-pods fill 2 GPUs; 4 Opportunistic request-0 pods
    still bind on the full GPUs and burst on gpu_limit (the reference's
    oversubscription path: guaranteed requests sum to <=1/GPU, while
    request-0 opportunistic pods pass Filter on any GPU and are
    time-sliced by L1; pod.go:276-305, filter.go:32-104)."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    g = [fc.add_pod("default", f"gu{i}",
                    shared("0.5", **{C.POD_PRIORITY: "100"}))
         for i in range(4)]
    fc.schedule_pending()
    o = [fc.add_pod("default", f"op{i}", {C.POD_GPU_LIMIT: "1.0"})
         for i in range(4)]
    fc.schedule_pending()
    assert all(p.phase == "Bound" for p in g), [p.phase for p in g]
    assert all(p.phase == "Bound" for p in o), [p.phase for p in o]


# -------------------------------------------------------- restart resync
def test_restart_resync_rebuilds_reservations():
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    pod = fc.add_pod("default", "r1", shared("0.5"))
    fc.schedule_pending()
    ann = dict(pod.annotations)

    # fresh scheduler instance (restart), same inventory
    fc2 = FakeCluster(nodes={"node-a": {"gpus": 2}})
    err = fc2.scheduler.resync_bound_pod(
        "default", "r1", pod.labels, ann, "node-a", uid=pod.uid)
    assert err is None
    leaf = fc2.scheduler.tree.leaf_by_uuid[ann[C.POD_GPU_UUID]]
    assert leaf.available == pytest.approx(0.5)
    # port re-masked: a new pod gets a different port
    p2 = fc2.add_pod("default", "r2", shared("0.5"))
    fc2.schedule_pending()
    assert p2.annotations[C.POD_MANAGER_PORT] != ann[C.POD_MANAGER_PORT]


# ------------------------------------------------------------ node health
def test_unhealthy_node_filtered():
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}, "node-b": {"gpus": 2}})
    fc.scheduler.set_node_health("node-a", False)
    pods = [fc.add_pod("default", f"hl{i}", shared("1.0"))
            for i in range(2)]
    fc.schedule_pending()
    assert all(p.phase == "Bound" for p in pods)
    assert all(p.node == "node-b" for p in pods)


def test_delete_pod_reclaims():
    fc = FakeCluster(nodes={"node-a": {"gpus": 1}})
    pod = fc.add_pod("default", "d1", shared("0.5"))
    fc.schedule_pending()
    fc.delete_pod(pod.key)
    leaf = fc.scheduler.tree.leaves_on_node("node-a")[0]
    assert leaf.available == 1.0
    assert leaf.free_memory == leaf.full_memory


def test_restart_resync_multi_gpu_pod():
    """Whole-GPU pods resync from their comma-joined uuid annotation
    (reference pod.go:348-400 + processBoundPod)."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 4}})
    pod = fc.add_pod("default", "mgr",
                     {C.POD_GPU_REQUEST: "2.0", C.POD_GPU_LIMIT: "2.0"})
    fc.schedule_pending()
    assert pod.phase == "Bound"

    fc2 = FakeCluster(nodes={"node-a": {"gpus": 4}})
    err = fc2.scheduler.resync_bound_pod(
        "default", "mgr", pod.labels, dict(pod.annotations), "node-a",
        uid=pod.uid)
    assert err is None
    used = [c for c in fc2.scheduler.tree.leaves_on_node("node-a")
            if c.available < 1.0]
    assert len(used) == 2
    assert all(c.available == 0.0 and c.free_memory == 0 for c in used)


def test_periodic_inventory_resync_preserves_reservations():
    """register_node is called periodically by the kube driver's
    sync_nodes; re-registering the same inventory must NOT wipe live
    reservations."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    pod = fc.add_pod("default", "keep", shared("0.5"))
    fc.schedule_pending()
    leaf = fc.scheduler.tree.leaf_by_uuid[pod.annotations[C.POD_GPU_UUID]]
    assert leaf.available == pytest.approx(0.5)
    fc.scheduler.register_node("node-a", fc.inventory.by_model("node-a"))
    assert leaf.available == pytest.approx(0.5), "resync wiped reservation"
    node_cell = fc.scheduler.tree.node_cells["node-a"][0]
    assert node_cell.available == pytest.approx(1.5)


def test_mixed_gpu_models_priority_and_pinning():
    """A cluster mixing two GPU models: higher childCellPriority model
    is preferred for unpinned pods (reference sortGPUPriority
    cell.go:57-72), and gpu_model pins to the right leaves."""
    from kubeshare_amd.scheduler.topology import CellSpec, CellTypeSpec
    topo = TopologyConfig(
        cell_types={
            "MI355X-NODE": CellTypeSpec("AMD Instinct MI355X", 2, 200, True),
            "MI300X-NODE": CellTypeSpec("AMD Instinct MI300X", 2, 100, True),
        },
        cells=[CellSpec(cell_type="MI355X-NODE", cell_id="fast-node"),
               CellSpec(cell_type="MI300X-NODE", cell_id="slow-node")])
    fc = FakeCluster(topology=topo,
                     nodes={"fast-node": {"gpus": 2},
                            "slow-node": {"gpus": 2,
                                          "model": "AMD Instinct MI300X",
                                          "memory": 192 * 1024**3}})
    assert fc.scheduler.tree.models_by_priority[0] == "AMD Instinct MI355X"

    # guarantee pod without a model pin lands on the faster model
    p = fc.add_pod("default", "fastp",
                   shared("0.5", **{C.POD_PRIORITY: "100"}))
    fc.schedule_pending()
    assert p.phase == "Bound" and p.node == "fast-node"

    # pinned to the slower model
    p2 = fc.add_pod("default", "slowp",
                    shared("0.5", **{C.POD_GPU_MODEL:
                                     "AMD Instinct MI300X"}))
    fc.schedule_pending()
    assert p2.phase == "Bound" and p2.node == "slow-node"
    # default memory derives from THAT model's capacity
    assert int(p2.annotations[C.POD_GPU_MEMORY]) == 96 * 1024**3


def test_multi_gpu_memory_constraint_respected_at_reserve():
    """A multi-GPU pod with a gpu_mem demand must only be placed on
    leaves satisfying it — also at cell-selection time, not just in
    Filter."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 4}})
    # shrink two leaves' free memory below the demand
    for c in fc.scheduler.tree.leaves_on_node("node-a")[:2]:
        fc.scheduler.tree.reserve(c, 0.0, C.MI355X_HBM_BYTES // 2)
    demand = str(C.MI355X_HBM_BYTES - 1024)
    pod = fc.add_pod("default", "mgm",
                     {C.POD_GPU_REQUEST: "2.0", C.POD_GPU_LIMIT: "2.0",
                      C.POD_GPU_MEMORY: demand})
    fc.schedule_pending()
    assert pod.phase == "Bound"
    picked = pod.annotations[C.POD_GPU_UUID].split(",")
    assert set(picked) <= {"GPU-node-a-2", "GPU-node-a-3"}, picked


def test_guarantee_pods_win_scarce_capacity():
    """QueueSort priority ordering: when capacity is scarce, Guarantee
    pods are scheduled before Opportunistic ones in the same cycle
    (reference Less, scheduler.go:247-267)."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    opps = [fc.add_pod("default", f"zo{i}", shared("1.0"))
            for i in range(4)]  # names sort AFTER guarantee pods anyway
    guas = [fc.add_pod("default", f"ag{i}",
                       shared("1.0", **{C.POD_PRIORITY: "100"}))
            for i in range(2)]
    fc.schedule_pending()
    assert all(p.phase == "Bound" for p in guas), [p.phase for p in guas]
    assert all(p.phase != "Bound" for p in opps)


def test_reserve_reclaim_float_drift():
    """Fractional reserve/reclaim cycles must restore EXACT whole-GPU
    availability (1.0 - 0.3 - 0.1 + 0.1 + 0.3 = 0.999... in doubles
    would otherwise permanently leak whole-GPU capacity)."""
    import itertools
    fc = FakeCluster(nodes={"node-a": {"gpus": 1}})
    tree = fc.scheduler.tree
    leaf = tree.leaves_on_node("node-a")[0]
    fracs = [0.1, 0.2, 0.25, 0.3, 0.33, 0.4, 0.5, 0.75]
    for combo in itertools.combinations_with_replacement(fracs, 2):
        if sum(combo) > 1.0:
            continue
        for r in combo:
            tree.reserve(leaf, r, 0)
        for r in combo:
            tree.reclaim(leaf, r, 0)
        assert leaf.available == 1.0, (combo, leaf.available)
        assert leaf.available_whole == 1, combo
    # a whole-GPU pod still fits after heavy fractional churn
    pod = fc.add_pod("default", "whole",
                     {C.POD_GPU_REQUEST: "1.0", C.POD_GPU_LIMIT: "1.0"})
    fc.schedule_pending()
    assert pod.phase == "Bound"
