"""Round-trip + cross-language tests of the per-UUID config file
contract (reference pkg/config/query.go:70-105): the Python writer must
be parsed identically by the C++ gpu-schd (verified through STATS)."""
import json
import socket
import subprocess
import time

from kubeshare_amd.configdaemon import files as F


def test_roundtrip_gpu_config(tmp_path):
    quotas = [F.PodQuota("ns/a", 1.0, 0.5, 144 * 2**30),
              F.PodQuota("ns/b", 0.25, 0.25, 0)]
    path = F.write_gpu_config(str(tmp_path), "GPU-42", quotas)
    assert F.read_gpu_config(path) == quotas


def test_roundtrip_port_config(tmp_path):
    ports = [F.PodPort("ns/a", 50050), F.PodPort("ns/b", 50051)]
    path = F.write_port_config(str(tmp_path), "GPU-42", ports)
    assert F.read_port_config(path) == ports


def test_zero_files(tmp_path):
    cfg = tmp_path / "c"
    prt = tmp_path / "p"
    cfg.mkdir()
    prt.mkdir()
    F.write_gpu_config(str(cfg), "GPU-1", [F.PodQuota("ns/a", 1.0, 0.5, 0)])
    F.write_port_config(str(prt), "GPU-1", [F.PodPort("ns/a", 50050)])
    F.zero_files(str(cfg), str(prt))
    assert (cfg / "GPU-1").read_text() == "0\n"
    assert (prt / "GPU-1").read_text() == "0\n"


def test_cpp_parses_python_writer(native_bins, tmp_path):
    cfg = tmp_path / "config"
    cfg.mkdir()
    F.write_gpu_config(str(cfg), "GPU-x",
                       [F.PodQuota("ns/px", 0.8, 0.6, 123456789)])
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [native_bins["gpu-schd"], "-p", str(cfg), "-f", "GPU-x",
         "-P", str(port)], stderr=subprocess.DEVNULL)
    try:
        deadline = time.time() + 5
        while time.time() < deadline:
            try:
                c = socket.create_connection(("127.0.0.1", port), timeout=0.2)
                break
            except OSError:
                time.sleep(0.05)
        c.sendall(b"STATS\n")
        st = json.loads(c.makefile().readline())
        c.close()
        pod = st["pods"]["ns/px"]
        assert pod["request"] == 0.6
        assert pod["limit"] == 0.8
    finally:
        proc.kill()
        proc.wait()


def test_lease_ms_roundtrip(tmp_path):
    """Latency-class field: q=<ms> survives write -> python read and is
    parsed by gpu-schd's C++ parser (covered via the loopback grant
    test); order-independent with the gang group field."""
    from kubeshare_amd.configdaemon import files as F
    quotas = [
        F.PodQuota("ns/svc", 1.0, 0.3, 0, lease_ms=25),
        F.PodQuota("ns/gang", 1.0, 0.5, 123, group="g1", lease_ms=50),
        F.PodQuota("ns/plain", 0.5, 0.25, 0),
    ]
    F.write_gpu_config(str(tmp_path), "GPU-q", quotas)
    text = (tmp_path / "GPU-q").read_text()
    assert " q=25" in text and "g1 q=50" in text
    back = F.read_gpu_config(str(tmp_path / "GPU-q"))
    assert back[0].lease_ms == 25 and back[0].group == ""
    assert back[1].lease_ms == 50 and back[1].group == "g1"
    assert back[2].lease_ms == 0
