"""L3/L2 control-plane tests + the full-stack CPU E2E: scheduler
placement -> aggregator demand -> config daemon files -> node launcher
-> gpu-schd/pod-mgr -> token client. The whole chain the reference could
only exercise on a live cluster runs here in-process."""
import json
import os
import socket
import time

import pytest
from prometheus_client import generate_latest
from prometheus_client.core import CollectorRegistry

from kubeshare_amd.aggregator import (GPURequirementCollector, PodDemand,
                                      demand_from_pod)
from kubeshare_amd.collector import GPUCapacityCollector
from kubeshare_amd.configdaemon import files as F
from kubeshare_amd.configdaemon.daemon import ConfigDaemon
from kubeshare_amd.noded.launcher import NodeDaemon
from kubeshare_amd.scheduler.harness import FakeCluster
from kubeshare_amd.scheduler.inventory import FakeInventory
from kubeshare_amd.utils import constants as C


class LocalProvider:
    def __init__(self, inv, node):
        self.inv, self.node = inv, node

    def local_gpus(self):
        return self.inv.gpus(self.node)


def test_collector_exports_gpu_capacity():
    inv = FakeInventory({"node-a": {"gpus": 2}})
    reg = CollectorRegistry()
    reg.register(GPUCapacityCollector("node-a", LocalProvider(inv, "node-a")))
    text = generate_latest(reg).decode()
    assert 'gpu_capacity{' in text
    assert 'uuid="GPU-node-a-0"' in text
    assert 'model="AMD-Instinct-MI355X"' in text  # spaces -> dashes
    assert f'memory="{C.MI355X_HBM_BYTES}"' in text
    assert 'xgmi_links="1"' in text


def test_aggregator_exports_gpu_requirement():
    fc = FakeCluster(nodes={"node-a": {"gpus": 1}})
    pod = fc.add_pod("default", "p1", {C.POD_GPU_REQUEST: "0.5",
                                       C.POD_GPU_LIMIT: "1.0"})
    fc.schedule_pending()

    def source():
        return [d for d in (demand_from_pod(p) for p in fc.pods.values())
                if d is not None]

    reg = CollectorRegistry()
    reg.register(GPURequirementCollector(source))
    text = generate_latest(reg).decode()
    assert 'gpu_requirement{' in text
    assert 'pod="p1"' in text
    assert 'request="0.5"' in text
    assert f'port="{pod.annotations[C.POD_MANAGER_PORT]}"' in text


def test_config_daemon_writes_per_uuid_files(tmp_path):
    cfg = tmp_path / "config"
    prt = tmp_path / "port"
    cfg.mkdir()
    prt.mkdir()
    daemon = ConfigDaemon("node-a", str(cfg), str(prt))
    daemon.update([
        PodDemand("ns", "a", "u1", "node-a", "GPU-x", 1.0, 0.5,
                  1024, 50050),
        PodDemand("ns", "b", "u2", "node-a", "GPU-x", 0.5, 0.25, 0, 50051),
        PodDemand("ns", "other-node", "u3", "node-b", "GPU-y", 1.0, 0.5,
                  0, 50052),
        PodDemand("ns", "whole", "u4", "node-a", "GPU-z", 4.0, 4.0,
                  0, 0),  # multi-GPU: skipped (no isolation layer)
    ])
    quotas = F.read_gpu_config(str(cfg / "GPU-x"))
    assert [q.pod for q in quotas] == ["ns/a", "ns/b"]
    assert not (cfg / "GPU-y").exists()
    assert not (cfg / "GPU-z").exists()
    # pod removal -> file zeroed
    daemon.update([PodDemand("ns", "b", "u2", "node-a", "GPU-x", 0.5,
                             0.25, 0, 50051)])
    assert len(F.read_gpu_config(str(cfg / "GPU-x"))) == 1
    daemon.update([])
    assert (cfg / "GPU-x").read_text() == "0\n"


@pytest.fixture
def fake_gpus():
    return [{"uuid": f"GPU-fake-{i}", "model": C.MI355X_MODEL,
             "memory": C.MI355X_HBM_BYTES, "index": i} for i in range(2)]


def _token_roundtrip(port: int, pod: str, timeout=10.0) -> float:
    from kubeshare_amd.isolation.client import TokenClient
    c = TokenClient("127.0.0.1", port, pod, timeout=timeout)
    quota = c.acquire()
    c.release(5.0)
    c.close()
    return quota


def test_node_daemon_full_chain(tmp_path, native_bins, fake_gpus):
    """Launcher starts gpu-schd per GPU; config daemon publishes a pod;
    launcher spawns its pod-mgr; a token round-trips through the chain;
    removing the pod kills the pod-mgr."""
    from kubeshare_amd.isolation.local import free_port
    base_port = free_port()
    nd = NodeDaemon(str(tmp_path), base_port=base_port, base_quota=50,
                    min_quota=10, window=2000, gpus=fake_gpus)
    nd.start()
    mgr_port = free_port()
    try:
        daemon = ConfigDaemon("node-a", nd.config_dir, nd.port_dir)
        daemon.update([PodDemand("ns", "p1", "u1", "node-a", "GPU-fake-0",
                                 1.0, 0.5, 0, mgr_port)])
        deadline = time.time() + 10
        while time.time() < deadline:
            nd.poll_once()
            try:
                socket.create_connection(("127.0.0.1", mgr_port),
                                         timeout=0.2).close()
                break
            except OSError:
                time.sleep(0.1)
        else:
            raise TimeoutError("pod-mgr never came up")
        quota = _token_roundtrip(mgr_port, "ignored")
        assert quota > 0
        # gpu-schd accounted it under the pod-mgr's stamped identity
        s = socket.create_connection(("127.0.0.1", base_port), timeout=5)
        s.sendall(b"STATS\n")
        st = json.loads(s.makefile().readline())
        s.close()
        assert st["pods"]["ns/p1"]["grants"] == 1
        # pod removed -> pod-mgr reaped
        daemon.update([])
        for _ in range(20):
            nd.poll_once()
            time.sleep(0.05)
        assert nd.sup["GPU-fake-0"].procs == {}
    finally:
        nd.stop()


def test_full_stack_e2e(tmp_path, native_bins, fake_gpus):
    """L4 -> L3 -> L2 -> L1 on CPU: schedule two 0.5 pods onto one GPU,
    flow their demand into the per-UUID files, and verify gpu-schd
    enforces the two quotas through the launcher-spawned pod-mgrs."""
    # L4: schedule
    fc = FakeCluster(nodes={"node-a": {"gpus": 1}})
    pods = [fc.add_pod("default", f"e2e{i}",
                       {C.POD_GPU_REQUEST: "0.5", C.POD_GPU_LIMIT: "1.0"})
            for i in range(2)]
    fc.schedule_pending()
    assert all(p.phase == "Bound" for p in pods)
    uuid = pods[0].annotations[C.POD_GPU_UUID]

    # L2 node daemon with inventory matching the scheduler's view
    gpus = [{"uuid": uuid, "model": C.MI355X_MODEL,
             "memory": C.MI355X_HBM_BYTES, "index": 0}]
    from kubeshare_amd.isolation.local import free_port
    base_port = free_port()
    nd = NodeDaemon(str(tmp_path), base_port=base_port, base_quota=40,
                    min_quota=10, window=2000, gpus=gpus)
    nd.start()
    try:
        # L3: aggregator demand -> config daemon -> files.
        # The scheduler assigns pool ports (50050+); under parallel test
        # runs another worker's ephemeral port can collide with that
        # fixed range, so remap each demand to a freshly probed port
        # (the pool-port plumbing itself is covered in test_scheduler).
        from dataclasses import replace
        from kubeshare_amd.isolation.local import free_port
        demands = [d for d in (demand_from_pod(p) for p in fc.pods.values())
                   if d is not None]
        remap = {d.port: free_port() for d in demands}
        demands = [replace(d, port=remap[d.port]) for d in demands]
        ConfigDaemon("node-a", nd.config_dir, nd.port_dir).update(demands)
        ports = {p.key: remap[int(p.annotations[C.POD_MANAGER_PORT])]
                 for p in pods}
        deadline = time.time() + 10
        while time.time() < deadline:
            nd.poll_once()
            try:
                for prt in ports.values():
                    socket.create_connection(("127.0.0.1", prt),
                                             timeout=0.2).close()
                break
            except OSError:
                time.sleep(0.1)
        # L1: both pods can acquire through their own pod-mgr
        for key, prt in ports.items():
            assert _token_roundtrip(prt, "whatever") > 0
        s = socket.create_connection(("127.0.0.1", base_port), timeout=5)
        s.sendall(b"STATS\n")
        st = json.loads(s.makefile().readline())
        s.close()
        for p in pods:
            assert st["pods"][p.key]["request"] == pytest.approx(0.5)
            assert st["pods"][p.key]["grants"] >= 1
    finally:
        nd.stop()


def test_simulator_smoke():
    """Trace-driven scheduler load test (reference test/simulator) on
    the in-memory harness: 150 jobs, bounded cycle time, no wedge."""
    import subprocess
    import sys as _sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [_sys.executable, "tools/simulator.py", "--jobs", "150",
         "--nodes", "2"],
        cwd=repo, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert out["jobs"] == 150
    assert out["pods_ever_bound"] == out["pods_submitted"]  # all served
    assert out["p99_cycle_ms"] < 50.0


def test_sharepod_conversion():
    from kubeshare_amd.sharepod import sharepod_to_pod
    obj = {
        "apiVersion": "sharedgpu.kubeshare.amd/v1", "kind": "SharePod",
        "metadata": {"name": "sp1", "namespace": "team", "uid": "u-1"},
        "spec": {
            "gpuRequest": "0.5", "gpuLimit": "1.0", "priority": "100",
            "groupName": "g", "groupHeadcount": "2", "groupThreshold": "1.0",
            "template": {"spec": {"containers": [
                {"name": "main", "image": "rocm/pytorch"}]}},
        },
    }
    pod = sharepod_to_pod(obj)
    assert pod["metadata"]["name"] == "sp1"
    assert pod["metadata"]["labels"][C.POD_GPU_REQUEST] == "0.5"
    assert pod["metadata"]["labels"][C.POD_GROUP_NAME] == "g"
    assert pod["spec"]["schedulerName"] == "kubeshare-scheduler"
    assert pod["metadata"]["ownerReferences"][0]["uid"] == "u-1"


def test_noded_metrics_exporter(tmp_path, native_bins, fake_gpus):
    """gpu-schd STATS surfaced as Prometheus metrics after real grants."""
    from kubeshare_amd.isolation.local import free_port
    base_port = free_port()
    nd = NodeDaemon(str(tmp_path), base_port=base_port, base_quota=50,
                    min_quota=10, window=2000, gpus=fake_gpus[:1])
    nd.start()
    mgr_port = free_port()
    try:
        ConfigDaemon("node-a", nd.config_dir, nd.port_dir).update(
            [PodDemand("ns", "m1", "u1", "node-a", "GPU-fake-0",
                       1.0, 0.5, 0, mgr_port)])
        deadline = time.time() + 10
        while time.time() < deadline:
            nd.poll_once()
            try:
                socket.create_connection(("127.0.0.1", mgr_port),
                                         timeout=0.2).close()
                break
            except OSError:
                time.sleep(0.1)
        _token_roundtrip(mgr_port, "x")
        from kubeshare_amd.noded.metrics import GpuSchdCollector
        reg = CollectorRegistry()
        reg.register(GpuSchdCollector(
            {"GPU-fake-0": ("127.0.0.1", base_port)}, "node-a"))
        text = generate_latest(reg).decode()
        assert 'gpu_pod_token_grants_total{node="node-a",pod="ns/m1",' \
               'uuid="GPU-fake-0"}' in text
        assert "gpu_pod_window_usage_ms" in text
        # round-2 accounting observability: sampler mode (0 on CPU
        # boxes — wall/RET fallback) and unattributed busy
        assert 'gpu_schd_busy_sampler{node="node-a",' \
               'uuid="GPU-fake-0"} 0.0' in text
        assert "gpu_schd_unattributed_busy_ms" in text
    finally:
        nd.stop()


def test_demand_from_pod_rejects_junk_numerics():
    """A hand-crafted pod with non-numeric gpu_mem/port/lease_ms must
    yield None, not crash the aggregator scrape."""
    class P:
        namespace, name, uid, node = "ns", "bad", "u", "n"
        labels = {C.POD_GPU_LIMIT: "1.0", C.POD_GPU_REQUEST: "0.5",
                  C.POD_LEASE_MS: "fast"}
        annotations = {C.POD_GPU_UUID: "GPU-x", C.POD_GPU_MEMORY: "lots"}
        env = {}

    assert demand_from_pod(P()) is None
    P.labels[C.POD_LEASE_MS] = "25"
    P.annotations[C.POD_GPU_MEMORY] = "1024"
    d = demand_from_pod(P())
    assert d is not None and d.lease_ms == 25 and d.memory == 1024


def test_noded_metrics_skips_garbled_endpoint():
    """A gpu-schd replying garbage (mid-restart truncation) must not
    fail the whole node scrape — that endpoint is skipped."""
    import threading
    from kubeshare_amd.noded.metrics import GpuSchdCollector

    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(4)
    port = srv.getsockname()[1]
    stop = threading.Event()

    def junk_server():
        srv.settimeout(0.2)
        while not stop.is_set():
            try:
                c, _ = srv.accept()
            except OSError:
                continue
            try:
                c.recv(256)
                c.sendall(b"not json {{{\n")
            finally:
                c.close()

    t = threading.Thread(target=junk_server, daemon=True)
    t.start()
    try:
        from kubeshare_amd.isolation.local import free_port
        reg = CollectorRegistry()
        reg.register(GpuSchdCollector(
            {"GPU-bad": ("127.0.0.1", port),
             "GPU-down": ("127.0.0.1", free_port())}, "node-a"))
        text = generate_latest(reg).decode()
        # scrape survives; families render with no samples for bad GPUs
        assert "gpu_pod_window_usage_ms" in text
        assert "GPU-bad" not in text and "GPU-down" not in text
    finally:
        stop.set()
        t.join(timeout=2)
        srv.close()


def test_pod_group_gc():
    import time as _t
    from kubeshare_amd.scheduler.pod_group import PodGroupRegistry
    reg = PodGroupRegistry(expiration_sec=0.1)
    reg.get_or_create("ns", "g1", 100, 2)
    assert "ns/g1" in reg.groups
    reg.gc(now=_t.time() + 1.0)
    assert "ns/g1" not in reg.groups


def test_kube_modules_importable():
    """The real-cluster drivers import without the kubernetes client
    (it is only required at instantiation)."""
    import kubeshare_amd.scheduler.kube  # noqa: F401
    import kubeshare_amd.sharepod  # noqa: F401
    import kubeshare_amd.queryip  # noqa: F401


def test_gpu_schd_crash_recovery(tmp_path, native_bins, fake_gpus):
    """Kill a gpu-schd mid-run: the launcher restarts it, pod-mgr is
    respawned, and a token round-trips again (restart-as-recovery, the
    failure-detection property SURVEY.md §5 tracks)."""
    from kubeshare_amd.isolation.local import free_port
    base_port = free_port()
    nd = NodeDaemon(str(tmp_path), base_port=base_port, base_quota=50,
                    min_quota=10, window=2000, gpus=fake_gpus[:1])
    nd.start()
    mgr_port = free_port()
    try:
        ConfigDaemon("node-a", nd.config_dir, nd.port_dir).update(
            [PodDemand("ns", "c1", "u1", "node-a", "GPU-fake-0",
                       1.0, 0.5, 0, mgr_port)])
        deadline = time.time() + 10
        while time.time() < deadline:
            nd.poll_once()
            try:
                socket.create_connection(("127.0.0.1", mgr_port),
                                         timeout=0.2).close()
                break
            except OSError:
                time.sleep(0.1)
        assert _token_roundtrip(mgr_port, "x") > 0
        # kill the scheduler daemon by its exact PID
        nd.schd["GPU-fake-0"].kill()
        nd.schd["GPU-fake-0"].wait()
        deadline = time.time() + 10
        ok = False
        while time.time() < deadline and not ok:
            nd.poll_once()
            try:
                if _token_roundtrip(mgr_port, "x", timeout=3.0) > 0:
                    ok = True
            except OSError:
                time.sleep(0.2)
        assert ok, "token path did not recover after gpu-schd crash"
    finally:
        nd.stop()


def test_launcher_pod_churn(tmp_path, native_bins, fake_gpus):
    """Rapid add/remove of sharing pods: the launcher must converge to
    the file state every time (no zombie pod-mgrs, no missed spawns)."""
    from kubeshare_amd.isolation.local import free_port
    base_port = free_port()
    nd = NodeDaemon(str(tmp_path), base_port=base_port, base_quota=40,
                    min_quota=10, window=1500, gpus=fake_gpus[:1])
    nd.start()
    daemon = ConfigDaemon("node-a", nd.config_dir, nd.port_dir)
    churn_ports = [free_port() for _ in range(3)]
    try:
        for cycle in range(6):
            pods = [PodDemand("ns", f"c{cycle}-{i}", f"u{i}", "node-a",
                              "GPU-fake-0", 1.0, 0.3, 0, churn_ports[i])
                    for i in range(cycle % 3 + 1)]
            daemon.update(pods)
            deadline = time.time() + 10
            want = {churn_ports[i] for i in range(cycle % 3 + 1)}
            while time.time() < deadline:
                nd.poll_once()
                if {nd.sup["GPU-fake-0"].ports[p]
                        for p in nd.sup["GPU-fake-0"].procs} == want:
                    break
                time.sleep(0.05)
            got = {nd.sup["GPU-fake-0"].ports[p]
                   for p in nd.sup["GPU-fake-0"].procs}
            assert got == want, f"cycle {cycle}: {got} != {want}"
        daemon.update([])
        for _ in range(40):
            nd.poll_once()
            if not nd.sup["GPU-fake-0"].procs:
                break
            time.sleep(0.05)
        assert nd.sup["GPU-fake-0"].procs == {}
    finally:
        nd.stop()


def test_pod_mgr_uds_default_transport(tmp_path, native_bins, fake_gpus):
    """pod-mgr listens on BOTH the per-pod UDS (default transport, no
    hostNetwork, unreachable cross-pod) and the TCP fallback; a token
    round-trips over each."""
    from kubeshare_amd.isolation.local import free_port
    base_port = free_port()
    nd = NodeDaemon(str(tmp_path), base_port=base_port, base_quota=40,
                    min_quota=10, window=2000, gpus=fake_gpus)
    nd.start()
    mgr_port = free_port()
    uds = C.pod_manager_uds(mgr_port, nd.sock_dir)
    try:
        ConfigDaemon("node-a", nd.config_dir, nd.port_dir).update(
            [PodDemand("ns", "udsy", "u1", "node-a", "GPU-fake-0",
                       1.0, 0.5, 0, mgr_port)])
        deadline = time.time() + 10
        while time.time() < deadline:
            nd.poll_once()
            if os.path.exists(uds):
                try:
                    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                    s.connect(uds)
                    s.close()
                    break
                except OSError:
                    pass
            time.sleep(0.1)
        else:
            raise TimeoutError("pod-mgr UDS never came up")
        # over UDS
        from kubeshare_amd.isolation.client import TokenClient
        c = TokenClient(uds, 0, "ignored")
        assert c.acquire() > 0
        c.release(5.0)
        c.close()
        # TCP fallback still works on the same pod-mgr
        assert _token_roundtrip(mgr_port, "ignored") > 0
    finally:
        nd.stop()


@pytest.mark.slow
@pytest.mark.timeout(1500)  # ~200 s here; headroom for slower CI boxes
def test_simulator_reference_trace_full():
    """The reference's actual 989-job arrival trace
    (test/simulator/trace.txt) replayed end to end against the
    in-memory pipeline — the load test the reference could only run on
    a live lab cluster. Skipped when the reference checkout is absent."""
    import subprocess
    import sys as _sys
    trace = "/root/reference/test/simulator/trace.txt"
    if not os.path.exists(trace):
        pytest.skip("reference trace not available")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [_sys.executable, "tools/simulator.py", "--trace", trace,
         "--nodes", "4"],
        cwd=repo, capture_output=True, text=True, timeout=1400)
    assert r.returncode == 0, r.stderr[-1500:]
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert out["jobs"] == 989
    # gang-heavy overload: most pods still bind. (No wall-clock cycle
    # assertion here: under xdist CPU contention cycle times scale with
    # the machine load — ~0.2 ms per pending pod uncontended; the smoke
    # test asserts timing on its small fixed workload instead.)
    assert out["pods_ever_bound"] > 1200
    assert out["never_bound_after_drain"] < 700
