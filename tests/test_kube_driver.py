"""KubeDriver (scheduler/kube.py) against a stubbed kubernetes client:
verifies node sync from labels, pod scheduling, and the shadow-pod
recreate with env injection — without a cluster or the kubernetes
package (a stub module is injected into sys.modules)."""
import sys
import types
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


def _stub_kubernetes(v1):
    mod = types.ModuleType("kubernetes")
    mod.client = types.SimpleNamespace(
        CoreV1Api=lambda: v1,
        V1EnvVar=lambda name, value: NS(name=name, value=value),
        V1VolumeMount=lambda name, mount_path: NS(name=name,
                                                  mount_path=mount_path),
        V1Volume=lambda name, host_path: NS(name=name, host_path=host_path),
        V1HostPathVolumeSource=lambda path: NS(path=path),
    )
    mod.config = types.SimpleNamespace(
        load_incluster_config=lambda: None,
        load_kube_config=lambda: None)
    mod.watch = types.SimpleNamespace()
    sys.modules["kubernetes"] = mod
    return mod


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
    _stub_kubernetes(v1)
    try:
        from kubeshare_amd.scheduler.kube import KubeDriver
        d = KubeDriver(TopologyConfig.single_node("node-a", gpus=2),
                       inventory=FakeInventory({"node-a": {"gpus": 2}}))
        d.waiting_pods = {}
        d.sync_nodes()
        yield d, v1
    finally:
        sys.modules.pop("kubernetes", None)


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
    env = {e.name: e.value for e in body.spec.containers[0].env}
    assert env[C.ENV_ROCR_VISIBLE_DEVICES] in ("0", "1")
    assert env[C.ENV_LD_PRELOAD] == C.HOOK_SO_PATH
    assert env[C.ENV_POD_NAME] == "ns/p1"
    assert int(env[C.ENV_GPU_MEM]) == C.MI355X_HBM_BYTES // 2
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
    env = {e.name: e.value for e in body.spec.containers[0].env}
    assert C.ENV_LD_PRELOAD not in env  # isolation bypass (pod.go:348-400)
    assert len(env[C.ENV_ROCR_VISIBLE_DEVICES].split(",")) == 2


def test_kube_driver_rejects_bad_labels(driver):
    d, v1 = driver
    pod = _fake_pod("ns", "bad", {C.POD_GPU_REQUEST: "0.5",
                                  C.POD_GPU_LIMIT: "0.2"})
    v1.pods["ns/bad"] = pod
    d.run_once()
    assert v1.created == [] and v1.deleted == []
