"""In-memory scheduling harness — the fixture layer the reference never
had: drives KubeShareScheduler through full scheduling
cycles (QueueSort -> PreFilter -> Filter -> Score -> Normalize ->
Reserve -> Permit -> Bind) against a fake cluster, with a simulated
clock for gang timeouts. Used by the unit tests and the trace-driven
load simulator (tools/simulator.py)."""
from __future__ import annotations

import itertools
from dataclasses import dataclass, field

from ..utils import constants as C
from .inventory import FakeInventory
from .plugin import KubeShareScheduler, Placement
from .topology import TopologyConfig


@dataclass
class FakePod:
    namespace: str
    name: str
    labels: dict = field(default_factory=dict)
    annotations: dict = field(default_factory=dict)
    env: dict = field(default_factory=dict)
    node: str = ""
    uid: str = ""
    phase: str = "Pending"   # Pending | Waiting | Bound | Unschedulable

    @property
    def key(self):
        return f"{self.namespace}/{self.name}"


class FakeCluster:
    """Nodes + pods + the binding side-effects the real control plane
    would apply (annotations/env injection at Reserve, reference
    pod.go:402-476 — modeled as in-place mutation rather than the
    reference's delete-and-recreate shadow pod, whose UID churn breaks
    owner references; SURVEY.md §7 phase 1.3)."""

    def __init__(self, topology: TopologyConfig | None = None,
                 nodes: dict | None = None):
        nodes = nodes or {"node-a": {"gpus": 8}}
        if topology is None:
            from .topology import CellSpec, CellTypeSpec
            topology = TopologyConfig(
                cell_types={"MI355X-NODE": CellTypeSpec(
                    C.MI355X_MODEL, max(n.get("gpus", 8)
                                        for n in nodes.values()),
                    100, True)},
                cells=[CellSpec(cell_type="MI355X-NODE", cell_id=n)
                       for n in nodes],
            )
        self.inventory = FakeInventory(nodes)
        self.scheduler = KubeShareScheduler(topology)
        for node in nodes:
            self.scheduler.register_node(node, self.inventory.by_model(node))
        self.pods: dict[str, FakePod] = {}
        self.clock = 0.0
        self._uid = itertools.count(1)
        self.events: list = []

    # ---------------------------------------------------------------- pods
    def add_pod(self, namespace: str, name: str, labels: dict) -> FakePod:
        pod = FakePod(namespace=namespace, name=name, labels=dict(labels),
                      uid=f"uid-{next(self._uid)}")
        self.pods[pod.key] = pod
        return pod

    def delete_pod(self, key: str):
        pod = self.pods.pop(key, None)
        if pod is not None:
            self.scheduler.delete_pod(pod.namespace, pod.name)

    # ------------------------------------------------------------ schedule
    def schedule_pending(self, rounds: int = 4):
        """Run scheduling cycles until quiescent."""
        for _ in range(rounds):
            progressed = False
            for pod in self._queue():
                out = self.schedule_one(pod)
                progressed |= out in ("Bound", "Waiting")
            self._expire_waiting()
            if not progressed:
                break
        return self

    def _queue(self):
        """Reference QueueSort order (scheduler.go:247-267): group
        priority desc, then enqueue time, then key. All pending pods
        share this simulated cycle's clock, so time degenerates to
        the key tiebreak."""
        pending = [p for p in self.pods.values() if p.phase == "Pending"]

        def sort_key(p):
            spec = self.scheduler.pod_status.get(p.key)
            prio = spec.priority if spec else 0
            return (-prio, p.key)
        return sorted(pending, key=sort_key)

    def schedule_one(self, pod: FakePod) -> str:
        sch = self.scheduler
        # gang totals come from the cluster's pod list (the reference
        # counts via its podLister, util.go:48-79), not from pod_status —
        # members not yet through PreFilter must still count
        group = pod.labels.get(C.POD_GROUP_NAME, "")
        in_group = sum(
            1 for p in self.pods.values()
            if p.namespace == pod.namespace
            and p.labels.get(C.POD_GROUP_NAME, "") == group) if group else None
        spec, err = sch.pre_filter(pod.namespace, pod.name, pod.labels,
                                   uid=pod.uid, all_pods_in_group=in_group)
        if spec is None:
            if err is None:
                pod.phase = "Regular"   # not a shared-GPU pod
            else:
                pod.phase = "Unschedulable"
                self.events.append(("prefilter-reject", pod.key, err))
            return pod.phase

        feasible = []
        for node in sch.tree.all_nodes():
            ok, msg = sch.filter(spec, node)
            if ok:
                feasible.append(node)
            else:
                self.events.append(("filter", pod.key, node, msg))
        if not feasible:
            pod.phase = "Unschedulable"
            return pod.phase

        scores = {n: sch.score(spec, n) for n in feasible}
        scores = sch.normalize_scores(scores)
        best = max(sorted(scores), key=lambda n: scores[n])

        placement = sch.reserve(spec, best)
        if placement is None:
            pod.phase = "Unschedulable"
            return pod.phase
        self._apply_injection(pod, placement)

        decision, timeout, release = sch.permit(spec, now=self.clock)
        if decision == "allow":
            self._bind(pod)
            for key in release:
                other = self.pods.get(key)
                if other is not None and other.phase == "Waiting":
                    self._bind(other)
            return "Bound"
        pod.phase = "Waiting"
        return "Waiting"

    def _apply_injection(self, pod: FakePod, placement: Placement):
        pod.annotations.update(placement.annotations)
        pod.env.update(placement.env)
        pod.node = placement.node

    def _bind(self, pod: FakePod):
        pod.phase = "Bound"
        self.events.append(("bind", pod.key, pod.node))

    def _expire_waiting(self):
        """Simulated clock pass: reject gangs whose Permit wait expired
        (reference Unreserve rejects all waiting members,
        scheduler.go:534-549)."""
        sch = self.scheduler
        for k in sch.expired_waiting(self.clock):
            pod = self.pods.get(k)
            spec = sch.pod_status.get(k)
            if spec is not None:
                sch.unreserve(spec)
            if pod is not None:
                pod.phase = "Unschedulable"
                for a in (C.POD_GPU_UUID, C.POD_CELL_ID,
                          C.POD_GPU_MEMORY, C.POD_MANAGER_PORT):
                    pod.annotations.pop(a, None)

    def advance(self, seconds: float):
        self.clock += seconds
        self._expire_waiting()
        return self

    # -------------------------------------------------------------- state
    def bound(self):
        return [p for p in self.pods.values() if p.phase == "Bound"]

    def leaf_state(self, node: str):
        return {c.id: (round(c.available, 4), c.free_memory)
                for c in self.scheduler.tree.leaves_on_node(node)}
