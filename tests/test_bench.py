"""bench.py contract tests on CPU (plumbing only: tiny model, fp32,
no GPU, gloo for the distributed path)."""
import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY = ["--model", "resnet18", "--batch", "2", "--image-size", "64",
        "--steps", "1", "--warmup", "0", "--device", "cpu",
        "--dtype", "fp32", "--use-ops", "off"]


def _parse_json_line(stdout: str) -> dict:
    for line in stdout.splitlines():
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout}")


def test_bench_single(native_bins):
    r = subprocess.run([sys.executable, "bench.py", "--gpus", "1"] + TINY,
                       cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    out = _parse_json_line(r.stdout)
    for key in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"]:
        assert key in out, key
    assert out["n_gpus"] == 1
    assert out["value"] > 0
    assert out["data"] == "synthetic"
    assert out["config"]["pods_per_gpu"] == 2
    assert out["config"]["gpu_request"] == 0.5


def test_bench_dist2_gloo(native_bins):
    port = socket.socket()
    port.bind(("127.0.0.1", 0))
    p = port.getsockname()[1]
    port.close()
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(p), "bench.py", "--gpus", "2"] + TINY,
        cwd=REPO, capture_output=True, text=True, timeout=900,
        env=dict(os.environ, MASTER_ADDR="127.0.0.1"))
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-3000:])
    out = _parse_json_line(r.stdout)
    assert out["n_gpus"] == 2
    # 2 ranks x 2 pods x 1 step x batch 2 images aggregated
    assert out["config"]["global_batch"] == 8


def test_bench_dist8_gloo(native_bins):
    """8-rank readiness on CPU: the first 8-GPU driver
    run must not be burned on plumbing. 8 ranks x 2 pods = 16 worker
    processes + 8 gpu-schd + 16 pod-mgr over deterministic UDS paths."""
    port = socket.socket()
    port.bind(("127.0.0.1", 0))
    p = port.getsockname()[1]
    port.close()
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", str(p), "bench.py", "--gpus", "8"] + TINY,
        cwd=REPO, capture_output=True, text=True, timeout=1800,
        env=dict(os.environ, MASTER_ADDR="127.0.0.1"))
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-3000:])
    out = _parse_json_line(r.stdout)
    assert out["n_gpus"] == 8
    assert out["config"]["global_batch"] == 32  # 8 x 2 pods x batch 2
    assert out["value"] > 0
