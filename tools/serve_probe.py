"""Serving-under-sharing probe: a hipGraph-replay inference pod
co-located with a saturating training pod on one MI355X, through the
full isolation chain. Reports inference tail latency + the trainer's
throughput cost as a function of the scheduler's base quota -q
(the latency/throughput tradeoff documented in BASELINE.md).

    gpurun -- 'python tools/serve_probe.py --quota-ms 300 --quota-ms 50'
"""
import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def serve_worker():
    import torch

    from kubeshare_amd.models import build_model
    from kubeshare_amd.serving import GraphReplayServer
    from kubeshare_amd.utils.tuning import apply_miopen_tuning

    apply_miopen_tuning()
    model = build_model("resnet50").to("cuda").to(
        memory_format=torch.channels_last).to(torch.bfloat16)
    x = torch.randn(8, 3, 224, 224, device="cuda",
                    dtype=torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    srv = GraphReplayServer(model, x)
    srv.serve(x)  # one gated end-to-end request before measuring
    print("READY", flush=True)
    sys.stdin.readline()
    out = srv.bench(x, n_requests=int(os.environ.get("SERVE_N", "120")),
                    interarrival_s=float(os.environ.get("SERVE_GAP",
                                                        "0.05")))
    print("SERVED " + json.dumps(out), flush=True)


def run_config(quota_ms: float, duration_ms: float) -> dict:
    from kubeshare_amd.isolation.local import LocalGPUShare
    share = LocalGPUShare(gpu_index=0, base_quota_ms=quota_ms,
                          min_quota_ms=10, window_ms=4000)
    share.start()
    try:
        trainer = share.add_pod("mix/train", request=0.7, limit=1.0)
        server = share.add_pod("mix/serve", request=0.3, limit=1.0)
        env_t = trainer.env(gpu_index=0)
        env_t["PYTHONPATH"] = REPO + os.pathsep + env_t.get("PYTHONPATH", "")
        env_s = server.env(gpu_index=0)
        env_s["PYTHONPATH"] = REPO + os.pathsep + env_s.get("PYTHONPATH", "")
        pt = subprocess.Popen(
            [sys.executable, "-m", "kubeshare_amd.isolation.burn_worker",
             "--duration-ms", str(duration_ms), "--wait-go"],
            env=env_t, cwd=REPO, stdin=subprocess.PIPE,
            stdout=subprocess.PIPE, text=True, bufsize=1)
        ps = subprocess.Popen([sys.executable, __file__, "--worker"],
                              env=env_s, cwd=REPO, stdin=subprocess.PIPE,
                              stdout=subprocess.PIPE, text=True, bufsize=1)
        for p in (pt, ps):
            line = p.stdout.readline().strip()
            assert line == "READY", line
        for p in (pt, ps):
            p.stdin.write("GO\n")
            p.stdin.flush()
        out_s, _ = ps.communicate(timeout=duration_ms / 1000 + 240)
        out_t, _ = pt.communicate(timeout=240)
        st = share.stats()
        served = json.loads(
            [ln for ln in out_s.splitlines()
             if ln.startswith("SERVED")][-1][len("SERVED "):])
        trainer_used = float(out_t.split()[4])
        wall = float(out_t.split()[1])
        return {"quota_ms": quota_ms, "latency": served,
                "trainer_busy_frac": round(trainer_used / (wall * 1000), 3),
                "revokes": st.get("revokes"),
                "last_revoked": st.get("last_revoked"),
                "schd": {k: round(v["busy_share"], 3)
                         for k, v in st.get("pods", {}).items()}}
    finally:
        share.stop()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quota-ms", type=float, action="append", default=[])
    ap.add_argument("--duration-ms", type=float, default=9000)
    ap.add_argument("--worker", action="store_true")
    args = ap.parse_args()
    if args.worker:
        return serve_worker()
    for q in (args.quota_ms or [300.0, 50.0]):
        print(json.dumps(run_config(q, args.duration_ms)), flush=True)


if __name__ == "__main__":
    main()
