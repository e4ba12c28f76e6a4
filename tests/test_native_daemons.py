"""CPU tests of the native isolation chain: gpu-schd token policy over
the real wire protocol, pod-mgr identity stamping/relay, and the
hook-side client state machine (hook_selftest models a GPU-bound pod:
its lease wall time stands in for exclusive GPU occupancy).

This is the loopback fixture layer SURVEY.md §4 calls for — the
reference has no equivalent (its Gemini chain was only testable on a
live cluster).
"""
import json
import os
import socket
import subprocess
import time

import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _write_config(tmp_path, pods):
    cfg = tmp_path / "config"
    cfg.mkdir(exist_ok=True)
    lines = [f"{len(pods)}"] + [
        f"{pod} {limit} {request} {mem}" for pod, limit, request, mem in pods
    ]
    (cfg / "GPU-x").write_text("\n".join(lines) + "\n")
    return str(cfg)


class Schd:
    def __init__(self, native_bins, cfg_dir, *, q=60, m=10, w=3000):
        self.port = _free_port()
        self.proc = subprocess.Popen(
            [native_bins["gpu-schd"], "-p", cfg_dir, "-f", "GPU-x",
             "-P", str(self.port), "-q", str(q), "-m", str(m), "-w", str(w)],
            stderr=subprocess.DEVNULL)
        _wait_listening(self.port)

    def stats(self):
        s = socket.create_connection(("127.0.0.1", self.port), timeout=5)
        s.sendall(b"STATS\n")
        f = s.makefile()
        line = f.readline()
        s.close()
        return json.loads(line)

    def stop(self):
        self.proc.kill()
        self.proc.wait()


def _wait_listening(port, timeout=5.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            socket.create_connection(("127.0.0.1", port), timeout=0.2).close()
            return
        except OSError:
            time.sleep(0.05)
    raise TimeoutError(f"port {port} never came up")


def _run_pods(native_bins, port, pods, duration_ms):
    procs = [
        subprocess.Popen(
            [native_bins["hook_selftest"], "127.0.0.1", str(port), pod,
             str(duration_ms)],
            stdout=subprocess.PIPE, text=True)
        for pod in pods
    ]
    out = {}
    for pod, p in zip(pods, procs):
        stdout, _ = p.communicate(timeout=duration_ms / 1000 + 30)
        assert p.returncode == 0, f"{pod} failed"
        _, name, leases, granted = stdout.split()
        out[name] = (int(leases), float(granted))
    return out


def test_equal_split_two_pods(native_bins, tmp_path):
    cfg = _write_config(tmp_path, [("ns/a", 1.0, 0.5, 0),
                                   ("ns/b", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg)
    try:
        res = _run_pods(native_bins, schd.port, ["ns/a", "ns/b"], 3000)
        total = res["ns/a"][1] + res["ns/b"][1]
        share_a = res["ns/a"][1] / total
        # request 0.5 each: busy-time split 50/50 within one base quota
        # (tolerance covers CI-box scheduling jitter)
        assert abs(share_a - 0.5) < 0.10, res
        st = schd.stats()
        assert abs(st["pods"]["ns/a"]["busy_share"] - 0.5) < 0.10
    finally:
        schd.stop()


def test_asymmetric_requests(native_bins, tmp_path):
    cfg = _write_config(tmp_path, [("ns/big", 0.75, 0.75, 0),
                                   ("ns/small", 0.25, 0.25, 0)])
    schd = Schd(native_bins, cfg)
    try:
        res = _run_pods(native_bins, schd.port, ["ns/big", "ns/small"], 3000)
        total = res["ns/big"][1] + res["ns/small"][1]
        share_big = res["ns/big"][1] / total
        # hard limits equal to requests: the split must track 75/25
        assert abs(share_big - 0.75) < 0.12, res
    finally:
        schd.stop()


def test_work_conserving_burst(native_bins, tmp_path):
    """A pod with request 0.3 / limit 1.0 alone on the GPU gets ~all of
    it (the reference's work-conserving request->limit contract)."""
    cfg = _write_config(tmp_path, [("ns/solo", 1.0, 0.3, 0)])
    schd = Schd(native_bins, cfg)
    try:
        res = _run_pods(native_bins, schd.port, ["ns/solo"], 2000)
        # ~2000ms of wall time granted to the only client
        assert res["ns/solo"][1] > 1700, res
    finally:
        schd.stop()


def test_hard_limit_caps_usage(native_bins, tmp_path):
    """limit < 1.0 with no competitor: usage is capped at limit*window
    (the pod must NOT be able to burn 100%)."""
    cfg = _write_config(tmp_path, [("ns/capped", 0.4, 0.2, 0)])
    schd = Schd(native_bins, cfg, q=60, m=10, w=1500)
    try:
        res = _run_pods(native_bins, schd.port, ["ns/capped"], 3000)
        frac = res["ns/capped"][1] / 3000.0
        assert frac < 0.55, f"hard cap violated: {frac}"
        assert frac > 0.25, f"over-throttled: {frac}"
    finally:
        schd.stop()


def test_unknown_pod_runs_opportunistically(native_bins, tmp_path):
    """Config file may lag pod start; an unlisted pod must still make
    progress (request 0 / limit 1 defaults)."""
    cfg = _write_config(tmp_path, [("ns/known", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg)
    try:
        res = _run_pods(native_bins, schd.port, ["ns/ghost"], 1000)
        assert res["ns/ghost"][1] > 700
    finally:
        schd.stop()


def test_config_hot_reload(native_bins, tmp_path):
    cfg = _write_config(tmp_path, [("ns/a", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg)
    try:
        _write_config(tmp_path, [("ns/a", 0.9, 0.6, 0),
                                 ("ns/b", 0.9, 0.4, 0)])
        time.sleep(0.5)  # inotify turnaround
        res = _run_pods(native_bins, schd.port, ["ns/a", "ns/b"], 3000)
        total = res["ns/a"][1] + res["ns/b"][1]
        share_a = res["ns/a"][1] / total
        assert abs(share_a - 0.6) < 0.12, res
        st = schd.stats()
        assert st["pods"]["ns/a"]["request"] == pytest.approx(0.6)
    finally:
        schd.stop()


def test_pod_mgr_stamps_identity(native_bins, tmp_path):
    """The hook's claimed pod name must be overridden by pod-mgr's env
    (a lying container cannot appropriate another pod's quota)."""
    cfg = _write_config(tmp_path, [("ns/honest", 1.0, 0.5, 0),
                                   ("ns/victim", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg)
    mgr_port = _free_port()
    env = dict(os.environ,
               SCHEDULER_IP="127.0.0.1", SCHEDULER_PORT=str(schd.port),
               POD_MANAGER_PORT=str(mgr_port), POD_NAME="ns/honest")
    mgr = subprocess.Popen([native_bins["pod-mgr"]], env=env,
                           stderr=subprocess.DEVNULL)
    try:
        _wait_listening(mgr_port)
        # client CLAIMS to be ns/victim, connects through the manager
        res = _run_pods(native_bins, mgr_port, ["ns/victim"], 1500)
        assert res["ns/victim"][1] > 1000  # it did run...
        st = schd.stats()
        # ...but was accounted as ns/honest upstream
        assert st["pods"]["ns/honest"]["grants"] > 0
        assert st["pods"]["ns/victim"]["grants"] == 0
    finally:
        mgr.kill()
        mgr.wait()
        schd.stop()


def test_pod_mgr_concurrent_clients(native_bins, tmp_path):
    """Two processes inside one pod share its quota through one pod-mgr
    while a second pod (direct) competes: pod-level split stays 50/50."""
    cfg = _write_config(tmp_path, [("ns/multi", 1.0, 0.5, 0),
                                   ("ns/other", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg)
    mgr_port = _free_port()
    env = dict(os.environ,
               SCHEDULER_IP="127.0.0.1", SCHEDULER_PORT=str(schd.port),
               POD_MANAGER_PORT=str(mgr_port), POD_NAME="ns/multi")
    mgr = subprocess.Popen([native_bins["pod-mgr"]], env=env,
                           stderr=subprocess.DEVNULL)
    try:
        _wait_listening(mgr_port)
        procs = [
            subprocess.Popen([native_bins["hook_selftest"], "127.0.0.1",
                              str(mgr_port), "proc%d" % i, "3000"],
                             stdout=subprocess.PIPE, text=True)
            for i in range(2)
        ] + [
            subprocess.Popen([native_bins["hook_selftest"], "127.0.0.1",
                              str(schd.port), "ns/other", "3000"],
                             stdout=subprocess.PIPE, text=True)
        ]
        for p in procs:
            p.communicate(timeout=40)
            assert p.returncode == 0
        st = schd.stats()
        share_multi = st["pods"]["ns/multi"]["busy_share"]
        assert abs(share_multi - 0.5) < 0.12, st
    finally:
        mgr.kill()
        mgr.wait()
        schd.stop()


def test_dead_holder_is_revoked(native_bins, tmp_path):
    """A client killed while holding the token must not wedge the GPU:
    the next pod gets granted after the liveness revoke."""
    cfg = _write_config(tmp_path, [("ns/a", 1.0, 0.5, 0),
                                   ("ns/b", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg, q=50, m=10, w=2000)
    try:
        # a grabs a token then dies without RET (connection stays open
        # from schd's view until killed -> closed; use SIGKILL)
        a = subprocess.Popen([native_bins["hook_selftest"], "127.0.0.1",
                              str(schd.port), "ns/a", "60000"],
                             stdout=subprocess.DEVNULL)
        time.sleep(0.3)
        a.kill()
        a.wait()
        res = _run_pods(native_bins, schd.port, ["ns/b"], 1000)
        assert res["ns/b"][1] > 500, "token not released after holder death"
    finally:
        schd.stop()


def test_idle_release_work_conserving(native_bins, tmp_path):
    """A bursty pod (100ms active / 400ms idle) must hand the GPU to a
    continuous competitor during its idle phases via the hook watchdog,
    not stall it until the liveness revoke."""
    cfg = _write_config(tmp_path, [("ns/bursty", 1.0, 0.5, 0),
                                   ("ns/greedy", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg, q=150, m=10, w=4000)
    try:
        pa = subprocess.Popen(
            [native_bins["hook_selftest"], "127.0.0.1", str(schd.port),
             "ns/bursty", "4000", "gate", "100", "400"],
            stdout=subprocess.PIPE, text=True)
        pb = subprocess.Popen(
            [native_bins["hook_selftest"], "127.0.0.1", str(schd.port),
             "ns/greedy", "4000", "gate", "4000", "0"],
            stdout=subprocess.PIPE, text=True)
        out_a, _ = pa.communicate(timeout=60)
        out_b, _ = pb.communicate(timeout=60)
        assert pa.returncode == 0 and pb.returncode == 0
        used_a = float(out_a.split()[3])
        used_b = float(out_b.split()[3])
        idle_rel = int(out_a.split()[4].split("=")[1])
        assert idle_rel >= 1, f"watchdog never released: {out_a}"
        # greedy pod picks up bursty's idle time (work conservation);
        # absolute thresholds stay loose — CI boxes jitter the sleeps
        assert used_b > 1.5 * used_a, (out_a, out_b)
        assert used_b > 1800, (out_a, out_b)
        # bursty still makes progress during its active phases
        assert used_a > 250, (out_a, out_b)
    finally:
        schd.stop()


def _write_config_grouped(tmp_path, pods):
    """pods: (pod, limit, request, mem, group)."""
    cfg = tmp_path / "config"
    cfg.mkdir(exist_ok=True)
    lines = [f"{len(pods)}"] + [
        f"{pod} {limit} {request} {mem} {group}".rstrip()
        for pod, limit, request, mem, group in pods
    ]
    (cfg / "GPU-x").write_text("\n".join(lines) + "\n")
    return str(cfg)


def test_gang_members_co_granted(native_bins, tmp_path):
    """Two pods of one gang group are granted CONCURRENTLY (a DDP
    collective in one rank must never spin on a token-starved peer),
    while a third ungrouped pod waits for the whole gang to drain."""
    cfg = _write_config_grouped(tmp_path, [
        ("ns/g0", 1.0, 0.4, 0, "ddp"),
        ("ns/g1", 1.0, 0.4, 0, "ddp"),
        ("ns/solo", 1.0, 0.2, 0, ""),
    ])
    schd = Schd(native_bins, cfg, q=200, m=10, w=4000)
    try:
        from kubeshare_amd.isolation.client import TokenClient
        a = TokenClient("127.0.0.1", schd.port, "ns/g0")
        b = TokenClient("127.0.0.1", schd.port, "ns/g1")
        qa = a.acquire()
        assert qa > 0
        # second gang member must be granted IMMEDIATELY while the
        # first still holds (co-grant), not after qa expires
        t0 = time.time()
        qb = b.acquire()
        assert qb > 0
        assert time.time() - t0 < 1.0, "gang member was serialized"
        st = schd.stats()
        # both counted as holders: usage still zero, both granted once
        assert st["pods"]["ns/g0"]["grants"] == 1
        assert st["pods"]["ns/g1"]["grants"] == 1
        a.release(50.0)
        b.release(50.0)
        a.close()
        b.close()
    finally:
        schd.stop()


def test_non_gang_waits_for_gang_drain(native_bins, tmp_path):
    cfg = _write_config_grouped(tmp_path, [
        ("ns/g0", 1.0, 0.4, 0, "ddp"),
        ("ns/g1", 1.0, 0.4, 0, "ddp"),
        ("ns/solo", 1.0, 0.2, 0, ""),
    ])
    schd = Schd(native_bins, cfg, q=200, m=10, w=4000)
    try:
        from kubeshare_amd.isolation.client import TokenClient
        a = TokenClient("127.0.0.1", schd.port, "ns/g0")
        b = TokenClient("127.0.0.1", schd.port, "ns/g1")
        s = TokenClient("127.0.0.1", schd.port, "ns/solo")
        a.acquire()
        b.acquire()

        import threading
        got = {}

        def solo_acquire():
            got["t0"] = time.time()
            got["quota"] = s.acquire()
            got["t1"] = time.time()

        th = threading.Thread(target=solo_acquire)
        th.start()
        time.sleep(0.4)
        assert "quota" not in got, "solo pod granted while gang held"
        a.release(100.0)
        time.sleep(0.3)
        assert "quota" not in got, "granted before the WHOLE gang drained"
        b.release(100.0)
        th.join(timeout=10)
        assert got.get("quota", 0) > 0
        for c in (a, b, s):
            c.close()
    finally:
        schd.stop()


def test_unix_domain_socket_chain(native_bins, tmp_path):
    """gpu-schd on a Unix socket + pod-mgr bridging UDS upstream to a
    TCP client port — the no-hostNetwork deployment mode (SURVEY.md §5:
    prefer UDS through the shared hostPath)."""
    cfg = _write_config(tmp_path, [("ns/u1", 1.0, 0.5, 0)])
    uds = str(tmp_path / "schd.sock")
    schd = subprocess.Popen(
        [native_bins["gpu-schd"], "-p", cfg, "-f", "GPU-x",
         "-U", uds, "-q", "60", "-m", "10", "-w", "2000"],
        stderr=subprocess.DEVNULL)
    mgr_port = _free_port()
    env = dict(os.environ, SCHEDULER_UDS=uds,
               POD_MANAGER_PORT=str(mgr_port), POD_NAME="ns/u1")
    mgr = None
    try:
        deadline = time.time() + 5
        while time.time() < deadline and not os.path.exists(uds):
            time.sleep(0.05)
        mgr = subprocess.Popen([native_bins["pod-mgr"]], env=env,
                               stderr=subprocess.DEVNULL)
        _wait_listening(mgr_port)
        res = _run_pods(native_bins, mgr_port, ["claimed"], 800)
        assert res["claimed"][1] > 500
    finally:
        if mgr:
            mgr.kill()
            mgr.wait()
        schd.kill()
        schd.wait()


def test_schd_survives_client_churn(native_bins, tmp_path):
    """Soak: many short-lived clients, some killed mid-REQ, some
    disconnecting while holding the token — the scheduler must keep
    granting promptly afterwards (no wedged token, no stale-waiter
    leak)."""
    cfg = _write_config(tmp_path, [("ns/a", 1.0, 0.5, 0),
                                   ("ns/b", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg, q=40, m=10, w=1500)
    try:
        from kubeshare_amd.isolation.client import TokenClient
        import random
        rng = random.Random(0)
        for i in range(30):
            pod = rng.choice(["ns/a", "ns/b", "ns/ghost"])
            c = TokenClient("127.0.0.1", schd.port, pod)
            q = c.acquire()
            assert q > 0
            if rng.random() < 0.5:
                c.release(rng.uniform(1, 30))
            # else: vanish while holding -> force_release on disconnect
            c.close()
        # scheduler still healthy: a fresh client gets a token fast
        t0 = time.time()
        c = TokenClient("127.0.0.1", schd.port, "ns/a")
        assert c.acquire() > 0
        assert time.time() - t0 < 3.0
        c.release(5)
        c.close()
        st = schd.stats()
        assert st["pods"]["ns/a"]["grants"] >= 1
    finally:
        schd.stop()


def test_client_reconnects_after_schd_restart(native_bins, tmp_path):
    """The hook-side TokenClient reconnects transparently when gpu-schd
    restarts on the same port (SO_REUSEADDR + lazy reconnect)."""
    cfg = _write_config(tmp_path, [("ns/r", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg, q=50, m=10, w=2000)
    port = schd.port
    try:
        res = _run_pods(native_bins, port, ["ns/r"], 400)
        assert res["ns/r"][1] > 200
        schd.proc.kill()
        schd.proc.wait()
        # restart on the SAME port
        schd.proc = subprocess.Popen(
            [native_bins["gpu-schd"], "-p", cfg, "-f", "GPU-x",
             "-P", str(port), "-q", "50", "-m", "10", "-w", "2000"],
            stderr=subprocess.DEVNULL)
        _wait_listening(port)
        res = _run_pods(native_bins, port, ["ns/r"], 400)
        assert res["ns/r"][1] > 200
    finally:
        schd.stop()


def test_sched_unit(native_bins):
    """Deterministic virtual-clock unit tests of the C++ token policy
    (fairness, caps, decay, revocation, gang co-granting) — see
    native/schd/sched_test.cpp."""
    r = subprocess.run([native_bins["sched_test"]], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode == 0, r.stderr or r.stdout
    assert "sched_test OK" in r.stdout


def test_lease_class_hot_reload(native_bins, tmp_path):
    """Adding/removing q=<ms> in the per-UUID file mid-run re-sizes the
    pod's leases on the next grant (inotify reload, no restart)."""
    from kubeshare_amd.isolation.client import TokenClient
    cfg = _write_config(tmp_path, [("ns/svc", 1.0, 0.5, 0)])
    schd = Schd(native_bins, cfg)
    try:
        c = TokenClient("127.0.0.1", schd.port, "ns/svc")
        assert c.acquire() == 60.0           # base quota (-q 60)
        c.release(5.0)
        (tmp_path / "config" / "GPU-x").write_text(
            "1\nns/svc 1.0 0.5 0 q=15\n")
        time.sleep(0.5)                      # inotify turnaround
        assert c.acquire() == 15.0           # latency class applied
        c.release(5.0)
        (tmp_path / "config" / "GPU-x").write_text(
            "1\nns/svc 1.0 0.5 0\n")
        time.sleep(0.5)
        q = c.acquire()
        c.release(5.0)
        c.close()
        assert q == 60.0                     # override removed
    finally:
        schd.stop()
