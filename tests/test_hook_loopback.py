"""CPU loopback tests of the REAL libhiphook.so interposer against a
fake libamdhip64 (native/testlibs/): memory cap + hipMemGetInfo clamp,
token gating with drain-at-renewal, and the RCCL caller exemption —
the fake-GPU substrate SURVEY.md §7 ranks among the hardest parts the
reference never had (its Gemini hook was only testable on live GPUs).
"""
import json
import os
import socket
import subprocess
import time

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
NATIVE = os.path.join(REPO, "native")
TESTLIBS = os.path.join(NATIVE, "testlibs")


@pytest.fixture(scope="module")
def loopback(native_bins):
    targets = ["libamdhip64.so.7", "librccl.so.1", "hook_app"]
    # make resolves staleness itself (instant no-op when fresh)
    subprocess.run(["make", "-C", NATIVE, "testlibs"], check=True,
                   capture_output=True)
    return {t: os.path.join(TESTLIBS, t) for t in targets}


def _env(extra):
    env = dict(os.environ)
    env["LD_PRELOAD"] = os.path.join(NATIVE, "libhiphook.so")
    env["LD_LIBRARY_PATH"] = TESTLIBS + os.pathsep + \
        env.get("LD_LIBRARY_PATH", "")
    env.update(extra)
    return env


def test_memcap_through_real_interposer(loopback):
    r = subprocess.run(
        [loopback["hook_app"], "memcap"],
        env=_env({"KUBESHARE_GPU_MEM": str(1 << 30)}),
        capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, (r.returncode, r.stdout, r.stderr)
    assert "MEMCAP_OK" in r.stdout


def test_vmm_and_pitched_allocs_capped(loopback):
    """The expandable_segments path (hipMemCreate/hipMemRelease) and
    hipMallocPitch honor KUBESHARE_GPU_MEM (round-1 VERDICT Missing #5:
    these bypassed the cap entirely)."""
    r = subprocess.run(
        [loopback["hook_app"], "vmm"],
        env=_env({"KUBESHARE_GPU_MEM": str(1 << 30)}),
        capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, (r.returncode, r.stdout, r.stderr)
    assert "VMM_OK" in r.stdout


def _start_schd(native_bins, tmp_path, pods):
    cfg = tmp_path / "config"
    cfg.mkdir(exist_ok=True)
    lines = [f"{len(pods)}"] + [f"{p} {l} {q} 0" for p, l, q in pods]
    (cfg / "GPU-x").write_text("\n".join(lines) + "\n")
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [native_bins["gpu-schd"], "-p", str(cfg), "-f", "GPU-x",
         "-P", str(port), "-q", "60", "-m", "10", "-w", "3000"],
        stderr=subprocess.DEVNULL)
    deadline = time.time() + 5
    while time.time() < deadline:
        try:
            socket.create_connection(("127.0.0.1", port), timeout=0.2).close()
            break
        except OSError:
            time.sleep(0.05)
    return proc, port


def test_gate_through_real_interposer(native_bins, loopback, tmp_path):
    """The preloaded hook acquires leases from a live gpu-schd while
    the fake app 'computes'; renewals drain the fake backlog."""
    proc, port = _start_schd(native_bins, tmp_path,
                             [("lo/app", "1.0", "0.5")])
    try:
        r = subprocess.run(
            [loopback["hook_app"], "gate", "1200"],
            env=_env({"SCHEDULER_IP": "127.0.0.1",
                      "SCHEDULER_PORT": str(port),
                      "POD_NAME": "lo/app",
                      "KUBESHARE_REQUIRE_HOOK": "1"}),
            capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, (r.returncode, r.stdout, r.stderr)
        out = dict(kv.split("=") for kv in r.stdout.split()[1:])
        assert int(out["leases"]) >= 2, r.stdout     # renewals happened
        assert int(out["syncs"]) >= 1, r.stdout      # drain-at-renewal ran
        # server-side accounting saw the pod
        s = socket.create_connection(("127.0.0.1", port), timeout=5)
        s.sendall(b"STATS\n")
        st = json.loads(s.makefile().readline())
        s.close()
        assert st["pods"]["lo/app"]["grants"] >= 2
        assert st["pods"]["lo/app"]["total_used_ms"] > 200
    finally:
        proc.kill()
        proc.wait()


def test_rccl_exemption_through_real_interposer(native_bins, loopback,
                                                tmp_path):
    """Launch sites inside librccl.so.1 bypass the gate entirely: the
    collective runs without consuming a single lease."""
    proc, port = _start_schd(native_bins, tmp_path,
                             [("lo/rccl", "1.0", "0.5")])
    try:
        r = subprocess.run(
            [loopback["hook_app"], "rccl"],
            env=_env({"SCHEDULER_IP": "127.0.0.1",
                      "SCHEDULER_PORT": str(port),
                      "POD_NAME": "lo/rccl",
                      "KUBESHARE_REQUIRE_HOOK": "1"}),
            capture_output=True, text=True, timeout=60)
        assert r.returncode == 0, (r.returncode, r.stdout, r.stderr)
        assert "RCCL_OK" in r.stdout
    finally:
        proc.kill()
        proc.wait()


def test_scheduler_ip_file_fallback(native_bins, loopback, tmp_path):
    """Hook discovery parity with the reference's gemhook: when only
    POD_MANAGER_PORT is injected, the endpoint IP comes from
    schedulerIP.txt on the mounted library hostPath
    (cmd/kubeshare-query-ip/main.go:23-34) — round-1 advisor HIGH
    finding: without this the gate silently ran inert."""
    proc, port = _start_schd(native_bins, tmp_path,
                             [("lo/ipfile", "1.0", "0.5")])
    ipfile = tmp_path / "schedulerIP.txt"
    ipfile.write_text("127.0.0.1\n")
    try:
        r = subprocess.run(
            [loopback["hook_app"], "gate", "600"],
            env=_env({"POD_MANAGER_PORT": str(port),
                      "KUBESHARE_SCHEDULER_IP_FILE": str(ipfile),
                      "POD_NAME": "lo/ipfile",
                      "KUBESHARE_REQUIRE_HOOK": "1"}),
            capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, (r.returncode, r.stdout, r.stderr)
        out = dict(kv.split("=") for kv in r.stdout.split()[1:])
        assert int(out["leases"]) >= 1, r.stdout
    finally:
        proc.kill()
        proc.wait()


def test_latency_class_lease_override(native_bins, loopback, tmp_path):
    """A pod with q=<ms> in its per-UUID config line gets leases of
    that length (the sharedgpu/lease_ms latency class, end to end
    through the real gpu-schd + hook)."""
    cfg = tmp_path / "config"
    cfg.mkdir(exist_ok=True)
    (cfg / "GPU-x").write_text("1\nlo/svc 1.0 0.5 0 q=15\n")
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [native_bins["gpu-schd"], "-p", str(cfg), "-f", "GPU-x",
         "-P", str(port), "-q", "60", "-m", "10", "-w", "3000"],
        stderr=subprocess.DEVNULL)
    try:
        deadline = time.time() + 5
        while time.time() < deadline:
            try:
                socket.create_connection(("127.0.0.1", port),
                                         timeout=0.2).close()
                break
            except OSError:
                time.sleep(0.05)
        from kubeshare_amd.isolation.client import TokenClient
        c = TokenClient("127.0.0.1", port, "lo/svc")
        assert c.acquire(hint_ms=200.0) == 15.0  # override beats hint
        c.release(15.0)
        c.close()
    finally:
        proc.kill()
        proc.wait()
