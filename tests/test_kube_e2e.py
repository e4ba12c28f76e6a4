"""End-to-end control-plane test over real HTTP: KubeDriver + stdlib
REST client + in-process fake kube-apiserver + the real mutating
webhook (BASELINE config #1 — "one pod gpu_request=0.5 schedules on a
cluster with faked GPU inventory, plumbing only, no GPU").

The fake API server (kubeshare_amd.testing.fake_apiserver) serves the
CoreV1 subset over HTTP with JSON and runs the actual webhook mutator
on pod CREATE; its kubelet emulation resolves the downward-API hostIP.
This is the closest automated equivalent of the reference's manual
kind/lab-cluster install check that runs in CI.
"""
import pytest

from kubeshare_amd.scheduler.inventory import FakeInventory
from kubeshare_amd.scheduler.kube import KubeDriver
from kubeshare_amd.scheduler.kubeclient import RestCoreV1
from kubeshare_amd.scheduler.topology import TopologyConfig
from kubeshare_amd.testing.fake_apiserver import FakeAPIServer
from kubeshare_amd.utils import constants as C


@pytest.fixture
def cluster():
    srv = FakeAPIServer()
    srv.add_node("node-a", labels={"SharedGPU": "true"},
                 host_ip="10.1.2.3")
    port = srv.start()
    api = RestCoreV1(f"http://127.0.0.1:{port}")
    driver = KubeDriver(TopologyConfig.single_node("node-a", gpus=2),
                        inventory=FakeInventory({"node-a": {"gpus": 2}}),
                        api=api)
    driver.sync_nodes()
    try:
        yield srv, api, driver
    finally:
        srv.stop()


def test_shared_pod_schedules_end_to_end(cluster):
    srv, api, driver = cluster
    srv.submit_pod("default", "trainer",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0",
                    C.POD_PRIORITY: "100"})
    driver.run_once()

    # original deleted, shadow pod created pinned to the node
    assert ("default", "trainer") in srv.deleted
    pod = api.read_namespaced_pod("trainer", "default")
    assert pod.spec.node_name == "node-a"
    # kubelet emulation ran: Running, hostIP resolved
    assert pod.status.phase == "Running"
    env = {e["name"]: e.get("value")
           for e in pod.spec.containers[0].to_payload()["env"]}
    assert env[C.ENV_LD_PRELOAD] == C.HOOK_SO_PATH
    assert env[C.ENV_POD_NAME] == "default/trainer"
    # downward API fieldRef resolved to the NODE's IP, not 127.0.0.1
    assert env[C.ENV_POD_MANAGER_IP] == "10.1.2.3"
    assert env[C.ENV_POD_MANAGER_UDS].startswith(C.SOCK_DIR + "/")
    assert int(env[C.ENV_GPU_MEM]) == C.MI355X_HBM_BYTES // 2
    # annotations written by Reserve survived the HTTP round-trip
    ann = dict(pod.metadata.annotations.items())
    assert ann[C.POD_GPU_UUID].startswith("GPU-node-a-")
    assert ann[C.POD_MANAGER_PORT]
    # volumes: hook library (ro) + socket dir (rw)
    vols = {v["name"] for v in pod.spec.volumes}
    assert {"kubeshare-library", "kubeshare-sock"} <= vols
    # scheduler state charged
    leaf = driver.sched.tree.leaf_by_uuid[ann[C.POD_GPU_UUID]]
    assert leaf.available == pytest.approx(0.5)


def test_webhook_injects_on_annotated_create(cluster):
    """The shadow-pod-free flow: a pod created WITH the Reserve
    annotations but no env gets the full injection from the webhook
    running inside the API server's admission chain."""
    srv, api, driver = cluster
    pod = {
        "metadata": {"namespace": "default", "name": "hooked",
                     "annotations": {
                         C.POD_GPU_UUID: "GPU-node-a-1",
                         C.POD_CELL_ID: "node-a/1",
                         C.POD_GPU_MEMORY: "1000000",
                         C.POD_MANAGER_PORT: "50061",
                         C.POD_GPU_INDEX: "1"}},
        "spec": {"schedulerName": C.SCHEDULER_NAME, "nodeName": "node-a",
                 "containers": [{"name": "main", "image": "x",
                                 "env": [{"name": "ROCR_VISIBLE_DEVICES",
                                          "value": "0,1"}]}]},
        "status": {"phase": "Pending"},
    }
    api.create_namespaced_pod("default", pod)
    got = api.read_namespaced_pod("hooked", "default")
    env = {e["name"]: e.get("value")
           for e in got.spec.containers[0].to_payload()["env"]}
    # conflicting user-set visibility env was REPLACED, not honored
    assert env[C.ENV_ROCR_VISIBLE_DEVICES] == "1"
    assert env[C.ENV_LD_PRELOAD] == C.HOOK_SO_PATH
    assert env[C.ENV_POD_MANAGER_IP] == "10.1.2.3"
    assert env[C.ENV_INJECTED] == "1"


def test_gang_schedules_atomically_over_http(cluster):
    srv, api, driver = cluster
    labels = {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0",
              C.POD_GROUP_NAME: "ddp", C.POD_GROUP_HEADCOUNT: "2",
              C.POD_GROUP_THRESHOLD: "1.0"}
    srv.submit_pod("default", "rank0", labels)
    srv.submit_pod("default", "rank1", labels)
    driver.run_once()
    for name in ("rank0", "rank1"):
        pod = api.read_namespaced_pod(name, "default")
        assert pod.spec.node_name == "node-a", name
        assert pod.status.phase == "Running"


def test_resync_after_restart_over_http(cluster):
    """Scheduler restart recovery (reference pod.go:47-78, 528-617):
    a fresh driver rebuilds reservations from bound-pod annotations."""
    srv, api, driver = cluster
    srv.submit_pod("default", "t1",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
    driver.run_once()
    ann = dict(api.read_namespaced_pod("t1", "default")
               .metadata.annotations.items())

    fresh = KubeDriver(TopologyConfig.single_node("node-a", gpus=2),
                       inventory=FakeInventory({"node-a": {"gpus": 2}}),
                       api=api)
    fresh.sync_nodes()
    fresh.resync_bound()
    leaf = fresh.sched.tree.leaf_by_uuid[ann[C.POD_GPU_UUID]]
    assert leaf.available == pytest.approx(0.5)
    port = int(ann[C.POD_MANAGER_PORT])
    assert not fresh.sched.ports["node-a"].is_free(port)


def test_sharepod_crd_end_to_end(cluster):
    """SharePod CRD flow over HTTP: create a SharePod custom object ->
    controller materializes the labeled Pod -> scheduler places it ->
    controller mirrors phase/node into the SharePod status (reference
    ships the KubeShare 1.x CRD alongside the label API, crd/v1.yaml)."""
    from kubeshare_amd.scheduler.kubeclient import RestCustomObjects
    from kubeshare_amd.sharepod import SharePodController

    srv, api, driver = cluster
    crd = RestCustomObjects(api)
    G, V, P = SharePodController.GROUP, "v1", SharePodController.PLURAL
    crd.create_namespaced_custom_object(G, V, "default", P, {
        "apiVersion": f"{G}/v1", "kind": "SharePod",
        "metadata": {"name": "sp1", "namespace": "default"},
        "spec": {"gpuRequest": "0.5", "gpuLimit": "1.0", "priority": "50",
                 "template": {"spec": {"containers": [
                     {"name": "main", "image": "rocm/pytorch"}]}}},
    })
    ctl = SharePodController(v1=api, crd=crd)
    ctl.reconcile_once()           # materialize the pod
    pod = api.read_namespaced_pod("sp1", "default")
    assert pod.spec.scheduler_name == C.SCHEDULER_NAME
    labels = dict(pod.metadata.labels.items())
    assert labels[C.POD_GPU_REQUEST] == "0.5"
    assert labels[C.POD_PRIORITY] == "50"
    driver.run_once()              # schedule it
    ctl.reconcile_once()           # mirror status
    obj = crd.get_namespaced_custom_object(G, V, "default", P, "sp1")
    assert obj["status"]["phase"] == "Running"
    assert obj["status"]["node"] == "node-a"
    # steady state: further reconciles are read-only (no no-op status
    # writes from the 2s control loop)
    writes = []
    orig = crd.patch_namespaced_custom_object_status
    crd.patch_namespaced_custom_object_status = \
        lambda *a, **kw: writes.append(a) or orig(*a, **kw)
    ctl.reconcile_once()
    assert writes == []


def test_noded_publishes_inventory_annotation(cluster, tmp_path):
    """The node daemon closes the inventory loop: amdsmi inventory ->
    kubeshare.amd/gpus node annotation (with the xGMI link graph) ->
    KubeDriver.sync_nodes, no Prometheus round-trip."""
    from kubeshare_amd.noded.launcher import NodeDaemon
    from kubeshare_amd.scheduler.kube import KubeDriver
    from kubeshare_amd.scheduler.topology import TopologyConfig

    srv, api, driver = cluster
    srv.add_node("node-inv", ready=True)  # fresh unlabeled node
    gpus = [{"uuid": f"GPU-inv-{i}", "model": C.MI355X_MODEL,
             "memory": C.MI355X_HBM_BYTES, "index": i,
             "xgmi_links": {j: 1 for j in range(2) if j != i}}
            for i in range(2)]
    nd = NodeDaemon(str(tmp_path), gpus=gpus)
    ann = nd.publish_inventory(node_name="node-inv", api=api)
    assert "GPU-inv-0" in ann and "links=" in ann
    # a driver with NO injected provider now sees the node via the
    # annotation (SharedGPU=true label was patched too)
    topo = TopologyConfig.single_node("node-inv", gpus=2)
    d2 = KubeDriver(topo, inventory=None, api=api)
    d2.sync_nodes()
    leaf = d2.sched.tree.leaf_by_uuid["GPU-inv-0"]
    assert leaf.full_memory == C.MI355X_HBM_BYTES
    assert "GPU-inv-1" in (leaf.xgmi_peers or {})


def test_multi_node_health_and_gang_over_http(tmp_path):
    """Two-node cluster over HTTP: an unhealthy node is avoided, a
    4-member gang lands atomically on the healthy node, and a health
    recovery makes the second node usable again."""
    from kubeshare_amd.scheduler.topology import TopologyConfig
    srv = FakeAPIServer()
    srv.add_node("n1", labels={"SharedGPU": "true"}, host_ip="10.0.0.1")
    srv.add_node("n2", labels={"SharedGPU": "true"}, host_ip="10.0.0.2",
                 ready=False)
    port = srv.start()
    api = RestCoreV1(f"http://127.0.0.1:{port}")
    inv = FakeInventory({"n1": {"gpus": 4}, "n2": {"gpus": 4}})
    driver = KubeDriver(TopologyConfig.nodes(["n1", "n2"], gpus=4),
                        inventory=inv, api=api)
    driver.sync_nodes()
    try:
        labels = {C.POD_GPU_REQUEST: "1.0", C.POD_GPU_LIMIT: "1.0",
                  C.POD_GROUP_NAME: "g4", C.POD_GROUP_HEADCOUNT: "4",
                  C.POD_GROUP_THRESHOLD: "1.0"}
        for i in range(4):
            srv.submit_pod("default", f"g4-{i}", labels)
        driver.run_once()
        nodes = {api.read_namespaced_pod(f"g4-{i}",
                                         "default").spec.node_name
                 for i in range(4)}
        assert nodes == {"n1"}  # n2 unhealthy: whole gang on n1
        # n2 recovers; a new pod can land there (n1 is full)
        with srv.lock:
            srv.nodes["n2"]["status"]["conditions"][0]["status"] = "True"
        driver.sync_nodes()
        srv.submit_pod("default", "after",
                       {C.POD_GPU_REQUEST: "0.5",
                        C.POD_GPU_LIMIT: "1.0"})
        driver.run_once()
        pod = api.read_namespaced_pod("after", "default")
        assert pod.spec.node_name == "n2"
    finally:
        srv.stop()


def test_lease_class_reaches_config_file(cluster, tmp_path):
    """Latency-class chain over HTTP: sharedgpu/lease_ms label on a
    Running pod -> kube_pod_source -> ConfigDaemon -> per-UUID config
    line carries q=<ms> for gpu-schd."""
    from kubeshare_amd.aggregator.__main__ import kube_pod_source
    from kubeshare_amd.configdaemon import files as F
    from kubeshare_amd.configdaemon.daemon import ConfigDaemon

    srv, api, driver = cluster
    srv.submit_pod("default", "svc",
                   {C.POD_GPU_REQUEST: "0.3", C.POD_GPU_LIMIT: "1.0",
                    C.POD_LEASE_MS: "25"})
    driver.run_once()
    source = kube_pod_source(api=api)
    demands = [d for d in source() if d.node == "node-a"]
    assert demands and demands[0].lease_ms == 25
    cfg = tmp_path / "cfg"
    prt = tmp_path / "prt"
    cfg.mkdir()
    prt.mkdir()
    ConfigDaemon("node-a", str(cfg), str(prt)).update(demands)
    uuid = demands[0].uuid
    quotas = F.read_gpu_config(str(cfg / uuid))
    assert quotas[0].lease_ms == 25
    assert "q=25" in (cfg / uuid).read_text()


def test_restart_with_waiting_gang_leaks_nothing(cluster):
    """Driver crash while a partial gang is parked at Permit: the new
    driver (fresh process) rebuilds from BOUND pods only — the waiting
    members were never applied, so a clean resync shows no leaked
    reservations and the gang can complete later."""
    from kubeshare_amd.scheduler.kube import KubeDriver
    from kubeshare_amd.scheduler.topology import TopologyConfig
    srv, api, driver = cluster
    labels = {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0",
              C.POD_GROUP_NAME: "g2", C.POD_GROUP_HEADCOUNT: "2",
              C.POD_GROUP_THRESHOLD: "1.0"}
    srv.submit_pod("default", "g2-a", labels)
    # second member exists but is not schedulable this cycle
    pod_b = srv.submit_pod("default", "g2-b", dict(labels))
    with srv.lock:
        pod_b["status"]["phase"] = "Unknown"
    driver.run_once()
    assert "default/g2-a" in driver.waiting_pods   # parked, not applied

    # "crash": a fresh driver resyncs from the cluster
    fresh = KubeDriver(TopologyConfig.single_node("node-a", gpus=2),
                       inventory=FakeInventory({"node-a": {"gpus": 2}}),
                       api=api)
    fresh.sync_nodes()
    fresh.resync_bound()
    for leaf in fresh.sched.tree.leaves_on_node("node-a"):
        assert leaf.available == 1.0   # nothing leaked into the resync
    # the gang completes once both members are Pending
    with srv.lock:
        pod_b["status"]["phase"] = "Pending"
    fresh.run_once()
    for name in ("g2-a", "g2-b"):
        pod = api.read_namespaced_pod(name, "default")
        assert pod.spec.node_name == "node-a", name


def test_regular_pod_not_stranded(cluster):
    """A pod that names kubeshare-scheduler but has NO sharedgpu labels
    (the reference's 'regular pod', pod.go:303-305) must still get
    bound — inside kube-scheduler the default plugins would place it;
    the out-of-tree driver binds it least-loaded with no injection."""
    srv, api, driver = cluster
    srv.submit_pod("default", "plain", {})      # no labels at all
    driver.run_once()
    pod = api.read_namespaced_pod("plain", "default")
    assert pod.spec.node_name == "node-a"
    assert pod.status.phase == "Running"
    env = pod.spec.containers[0].to_payload().get("env")
    assert not env                              # no injection
    # and no scheduler-side GPU reservation was charged
    for leaf in driver.sched.tree.leaves_on_node("node-a"):
        assert leaf.available == 1.0


def test_apply_failure_rolls_back_reservation(cluster):
    """If a pod is deleted between Reserve and apply (shadow-pod
    recreate), the driver must unreserve — otherwise the next pod sees
    phantom usage and a retried namesake double-charges."""
    srv, api, driver = cluster
    srv.submit_pod("default", "ghost",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
    pods = driver.v1.list_pod_for_all_namespaces(
        field_selector="status.phase=Pending").items
    target = [p for p in pods if p.metadata.name == "ghost"][0]
    with srv.lock:
        srv.pods.pop(("default", "ghost"))   # user deletes it racily
    driver.schedule_pod(target)              # apply must fail + roll back
    for leaf in driver.sched.tree.leaves_on_node("node-a"):
        assert leaf.available == 1.0, leaf
    assert driver.sched.ports["node-a"].available()


def test_pod_merge_patch_route(cluster):
    srv, api, driver = cluster
    srv.submit_pod("default", "patchme", {})
    api.patch_namespaced_pod("patchme", "default", {
        "metadata": {"annotations": {"x": "1"}, "labels": {"y": "2"}}})
    pod = api.read_namespaced_pod("patchme", "default")
    assert pod.metadata.annotations["x"] == "1"
    assert pod.metadata.labels["y"] == "2"


def test_deleted_pod_reservations_reclaimed(cluster):
    """A long-running driver must reclaim when bound pods are deleted
    or complete (reference informer DeleteFunc, pod.go:91-136) — no
    restart required."""
    srv, api, driver = cluster
    srv.submit_pod("default", "d1",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
    driver.run_once()
    ann = dict(api.read_namespaced_pod("d1", "default")
               .metadata.annotations.items())
    leaf = driver.sched.tree.leaf_by_uuid[ann[C.POD_GPU_UUID]]
    port = int(ann[C.POD_MANAGER_PORT])
    assert leaf.available == pytest.approx(0.5)
    # user deletes the running pod
    api.delete_namespaced_pod("d1", "default")
    driver.run_once()
    assert leaf.available == pytest.approx(1.0)
    assert driver.sched.ports["node-a"].is_free(port)
    assert "default/d1" not in driver.sched.pod_status

    # completion (phase Succeeded) reclaims too
    srv.submit_pod("default", "d2",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
    driver.run_once()
    with srv.lock:
        srv.pods[("default", "d2")]["status"]["phase"] = "Succeeded"
    driver.run_once()
    for lf in driver.sched.tree.leaves_on_node("node-a"):
        assert lf.available == pytest.approx(1.0)


def test_namesake_recreation_reclaims_old_instance(cluster):
    """A controller deleting a bound pod and recreating the NAME with a
    new uid: the old instance's reservation is reclaimed before the new
    Pending instance schedules (no double-charge, no leak)."""
    srv, api, driver = cluster
    srv.submit_pod("default", "twin",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
    driver.run_once()
    # controller replaces it: same name, fresh uid, Pending, unbound
    with srv.lock:
        srv.pods.pop(("default", "twin"))
    srv.submit_pod("default", "twin",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
    driver.run_once()
    # exactly ONE 0.5 reservation exists afterwards
    total_reserved = sum(1.0 - lf.available
                         for lf in driver.sched.tree.leaves_on_node(
                             "node-a"))
    assert total_reserved == pytest.approx(0.5)
    pod = api.read_namespaced_pod("twin", "default")
    assert pod.spec.node_name == "node-a"


def test_unlabeled_node_stops_receiving_pods(cluster):
    """Removing SharedGPU=true (or the node leaving the cluster) makes
    its cells unhealthy on the next node sync (reference updateNode,
    node.go:54-68)."""
    srv, api, driver = cluster
    with srv.lock:
        srv.nodes["node-a"]["metadata"]["labels"].pop("SharedGPU")
    driver.sync_nodes()
    srv.submit_pod("default", "nohome",
                   {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
    driver.run_once()
    pod = api.read_namespaced_pod("nohome", "default")
    assert not pod.spec.node_name          # nowhere to go
    # relabel -> schedulable again
    with srv.lock:
        srv.nodes["node-a"]["metadata"]["labels"]["SharedGPU"] = "true"
    driver.sync_nodes()
    driver.run_once()
    pod = api.read_namespaced_pod("nohome", "default")
    assert pod.spec.node_name == "node-a"
