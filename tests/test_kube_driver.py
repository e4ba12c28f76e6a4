"""KubeDriver (scheduler/kube.py) against a stubbed kubernetes client:
verifies node sync from labels, pod scheduling, and the shadow-pod
recreate with env injection — without a cluster or the kubernetes
package (the client object is injected). The full HTTP round-trip
against the fake API server lives in test_kube_e2e.py."""
from types import SimpleNamespace as NS

import pytest

from kubeshare_amd.scheduler.inventory import FakeInventory
from kubeshare_amd.scheduler.topology import TopologyConfig
from kubeshare_amd.utils import constants as C


class FakeV1:
    def __init__(self):
        self.nodes = []
        self.pods = {}
        self.created = []
        self.deleted = []

    def list_node(self, label_selector=None):
        return NS(items=self.nodes)

    def list_pod_for_all_namespaces(self, field_selector=None):
        pods = list(self.pods.values())
        if field_selector == "status.phase=Pending":
            pods = [p for p in pods if p.status.phase == "Pending"]
        return NS(items=pods)

    def list_namespaced_pod(self, ns, label_selector=None):
        key, val = label_selector.split("=")
        return NS(items=[p for p in self.pods.values()
                         if (p.metadata.labels or {}).get(key) == val])

    def read_namespaced_pod(self, name, ns):
        return self.pods[f"{ns}/{name}"]

    def delete_namespaced_pod(self, name, ns, grace_period_seconds=None):
        self.deleted.append(f"{ns}/{name}")
        self.pods.pop(f"{ns}/{name}", None)

    def create_namespaced_pod(self, ns, body):
        self.created.append(body)
        self.pods[f"{ns}/{body.metadata.name}"] = body


def _fake_pod(ns, name, labels, uid="u1"):
    return NS(
        metadata=NS(namespace=ns, name=name, labels=labels,
                    annotations={}, uid=uid, resource_version="1",
                    creation_timestamp=1.0),
        spec=NS(scheduler_name=C.SCHEDULER_NAME, node_name=None,
                containers=[NS(env=None, volume_mounts=None)],
                volumes=None),
        status=NS(phase="Pending", conditions=[]),
    )


@pytest.fixture
def driver():
    v1 = FakeV1()
    v1.nodes = [NS(metadata=NS(name="node-a", annotations={}),
                   status=NS(conditions=[NS(type="Ready", status="True")]))]
    from kubeshare_amd.scheduler.kube import KubeDriver
    d = KubeDriver(TopologyConfig.single_node("node-a", gpus=2),
                   inventory=FakeInventory({"node-a": {"gpus": 2}}),
                   api=v1)
    d.sync_nodes()
    yield d, v1


def test_kube_driver_schedules_and_injects(driver):
    d, v1 = driver
    pod = _fake_pod("ns", "p1", {C.POD_GPU_REQUEST: "0.5",
                                 C.POD_GPU_LIMIT: "1.0"})
    v1.pods["ns/p1"] = pod
    d.run_once()
    # shadow-pod recreate: delete original, create injected copy
    assert v1.deleted == ["ns/p1"]
    assert len(v1.created) == 1
    body = v1.created[0]
    assert body.spec.node_name == "node-a"
    env = {e["name"]: e.get("value") for e in body.spec.containers[0].env}
    assert env[C.ENV_ROCR_VISIBLE_DEVICES] in ("0", "1")
    assert env[C.ENV_LD_PRELOAD] == C.HOOK_SO_PATH
    assert env[C.ENV_POD_NAME] == "ns/p1"
    assert int(env[C.ENV_GPU_MEM]) == C.MI355X_HBM_BYTES // 2
    # UDS default + node-IP via downward API (pod-mgr runs hostNetwork:
    # 127.0.0.1 inside the pod would be the pod itself)
    assert env[C.ENV_POD_MANAGER_UDS].startswith(C.SOCK_DIR + "/")
    by_name = {e["name"]: e for e in body.spec.containers[0].env}
    ref = by_name[C.ENV_POD_MANAGER_IP]["valueFrom"]["fieldRef"]
    assert ref["fieldPath"] == "status.hostIP"
    assert body.metadata.annotations[C.POD_GPU_UUID] == "GPU-node-a-0"
    assert body.metadata.annotations["kubeshare.amd/original-uid"] == "u1"
    # tree was charged
    leaf = d.sched.tree.leaf_by_uuid["GPU-node-a-0"]
    assert leaf.available == pytest.approx(0.5)


def test_kube_driver_whole_gpu_no_hook(driver):
    d, v1 = driver
    pod = _fake_pod("ns", "pw", {C.POD_GPU_REQUEST: "2.0",
                                 C.POD_GPU_LIMIT: "2.0"})
    v1.pods["ns/pw"] = pod
    d.run_once()
    body = v1.created[0]
    env = {e["name"]: e.get("value") for e in body.spec.containers[0].env}
    assert C.ENV_LD_PRELOAD not in env  # isolation bypass (pod.go:348-400)
    assert len(env[C.ENV_ROCR_VISIBLE_DEVICES].split(",")) == 2


def test_kube_driver_rejects_bad_labels(driver):
    d, v1 = driver
    pod = _fake_pod("ns", "bad", {C.POD_GPU_REQUEST: "0.5",
                                  C.POD_GPU_LIMIT: "0.2"})
    v1.pods["ns/bad"] = pod
    d.run_once()
    assert v1.created == [] and v1.deleted == []


def test_kube_driver_gang_permit_timeout_reclaims(driver):
    """A partial gang parked at Permit must not leak its reservations
    (reference scheduler.go:551-587 + Unreserve 534-549): after the
    2s x headcount deadline the driver unreserves the waiting members."""
    d, v1 = driver
    labels = {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0",
              C.POD_GROUP_NAME: "g1", C.POD_GROUP_HEADCOUNT: "2",
              C.POD_GROUP_THRESHOLD: "1.0"}
    v1.pods["ns/g1-a"] = _fake_pod("ns", "g1-a", labels)
    # PreFilter needs total group pods >= minAvailable: add the second
    # member to the cluster but keep it un-schedulable this cycle
    other = _fake_pod("ns", "g1-b", dict(labels), uid="u2")
    other.status.phase = "Unknown"
    v1.pods["ns/g1-b"] = other
    d.run_once()
    assert "ns/g1-a" in d.waiting_pods           # parked at Permit
    assert v1.created == []                      # not applied yet
    leaf = d.sched.tree.leaf_by_uuid["GPU-node-a-0"]
    assert leaf.available == pytest.approx(0.5)  # reserved
    # deadline passes (2s x headcount=2 -> 4s); force it
    for waiters in d.sched.waiting.values():
        for k in waiters:
            waiters[k] = 0.0
    d.expire_permits()
    assert d.waiting_pods == {}
    assert leaf.available == pytest.approx(1.0)  # reclaimed
    assert d.sched.pod_status["ns/g1-a"].port == 0
    # the pod is still Pending in the cluster: the next cycle retries it
    # (and parks it again while its gang stays incomplete)
    d.run_once()
    assert "ns/g1-a" in d.waiting_pods


def test_kube_driver_queue_order_priority_first(driver):
    """QueueSort parity (reference scheduler.go:247-267): a later-created
    Guarantee pod outranks an earlier Opportunistic pod when only one
    fits."""
    d, v1 = driver
    lo = _fake_pod("ns", "lo", {C.POD_GPU_REQUEST: "0.6",
                                C.POD_GPU_LIMIT: "1.0"}, uid="u-lo")
    hi = _fake_pod("ns", "hi", {C.POD_GPU_REQUEST: "0.6",
                                C.POD_GPU_LIMIT: "1.0",
                                C.POD_PRIORITY: "100"}, uid="u-hi")
    lo.metadata.creation_timestamp = 1.0
    hi.metadata.creation_timestamp = 2.0
    # shrink the cluster to one schedulable leaf
    for leaf in d.sched.tree.leaves_on_node("node-a")[1:]:
        d.sched.tree.reserve(leaf, 1.0, leaf.full_memory)
    v1.pods["ns/lo"] = lo
    v1.pods["ns/hi"] = hi
    d.run_once()
    created = {b.metadata.name for b in v1.created}
    assert "hi" in created
    assert "lo" not in created
