"""Hardware-level proof of quota enforcement: run 2 co-located pods
@0.5 through the full isolation chain with ONE of them under
rocprofv3's kernel trace, and compute that pod's GPU-busy fraction from
the hardware timestamps (sum of kernel durations / wall). Independent
of the scheduler's own accounting — the BASELINE.json evidence bar.

    gpurun -- 'python tools/quota_proof.py --duration-ms 8000'

Prints one JSON line: {"traced_busy_frac": ..., "expected": 0.5, ...}
"""
import argparse
import csv
import glob
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from kubeshare_amd.isolation.local import LocalGPUShare  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration-ms", type=float, default=8000)
    ap.add_argument("--request", type=float, default=0.5)
    ap.add_argument("--outdir", default="/tmp/quota_proof")
    args = ap.parse_args()

    share = LocalGPUShare(gpu_index=0, base_quota_ms=100, window_ms=4000)
    share.start()
    try:
        a = share.add_pod("proof/traced", request=args.request,
                          limit=args.request)
        b = share.add_pod("proof/other", request=1.0 - args.request,
                          limit=1.0 - args.request)
        os.makedirs(args.outdir, exist_ok=True)

        def worker_cmd():
            return [sys.executable, "-m",
                    "kubeshare_amd.isolation.burn_worker",
                    "--duration-ms", str(args.duration_ms), "--wait-go"]

        env_a = a.env(gpu_index=0)
        env_a["PYTHONPATH"] = REPO + os.pathsep + env_a.get("PYTHONPATH", "")
        env_b = b.env(gpu_index=0)
        env_b["PYTHONPATH"] = REPO + os.pathsep + env_b.get("PYTHONPATH", "")
        # rocprofv3 around pod A only; its kernel trace gives HW truth
        cmd_a = ["rocprofv3", "--kernel-trace", "--output-format", "csv",
                 "-d", args.outdir, "-o", "traced", "--"] + worker_cmd()
        pa = subprocess.Popen(cmd_a, env=env_a, cwd="/tmp",
                              stdin=subprocess.PIPE, stdout=subprocess.PIPE,
                              text=True, bufsize=1)
        pb = subprocess.Popen(worker_cmd(), env=env_b, cwd="/tmp",
                              stdin=subprocess.PIPE, stdout=subprocess.PIPE,
                              text=True, bufsize=1)
        for p in (pa, pb):
            line = p.stdout.readline().strip()
            assert line == "READY", line
        for p in (pa, pb):
            p.stdin.write("GO\n")
            p.stdin.flush()
        out_a, _ = pa.communicate(timeout=args.duration_ms / 1000 + 120)
        out_b, _ = pb.communicate(timeout=120)
        wall_s = args.duration_ms / 1000.0
        st = share.stats()
    finally:
        share.stop()

    # parse the kernel trace: columns include Start_Timestamp and
    # End_Timestamp (ns)
    busy_ns = 0
    tmin, tmax = None, None
    files = glob.glob(os.path.join(args.outdir, "**", "*kernel_trace*.csv"),
                      recursive=True)
    for path in files:
        with open(path) as f:
            for row in csv.DictReader(f):
                try:
                    s = int(row.get("Start_Timestamp") or
                            row.get("\"Start_Timestamp\"") or 0)
                    e = int(row.get("End_Timestamp") or 0)
                except (TypeError, ValueError):
                    continue
                if e <= s:
                    continue
                busy_ns += e - s
                tmin = s if tmin is None else min(tmin, s)
                tmax = e if tmax is None else max(tmax, e)

    span_s = (tmax - tmin) / 1e9 if tmin is not None else wall_s
    frac = busy_ns / 1e9 / max(span_s, 1e-9)
    print(json.dumps({
        "traced_busy_frac": round(frac, 4),
        "expected": args.request,
        "trace_span_s": round(span_s, 2),
        "kernel_busy_s": round(busy_ns / 1e9, 2),
        "trace_files": len(files),
        "schd_stats": {k: {"busy_share": round(v["busy_share"], 4),
                           "window_frac": round(v["window_frac"], 4)}
                       for k, v in st["pods"].items()},
        "worker_lines": [out_a.strip().splitlines()[-1] if out_a else "",
                         out_b.strip().splitlines()[-1] if out_b else ""],
    }))


if __name__ == "__main__":
    main()
