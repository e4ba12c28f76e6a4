"""Trace-driven scheduler load test (reference test/simulator/
simulator.py replays a 989-job trace of "start\\tgpus\\truntime" rows
against a live cluster via kubectl; this version replays the same trace
format against the in-memory harness, so the full QueueSort->Permit
pipeline is load-tested in CI with no cluster).

    python tools/simulator.py --jobs 500 --nodes 4
    python tools/simulator.py --trace /root/reference/test/simulator/trace.txt
"""
import argparse
import json
import os
import random
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from kubeshare_amd.scheduler.harness import FakeCluster  # noqa: E402
from kubeshare_amd.utils import constants as C  # noqa: E402


def synthetic_trace(jobs: int, seed: int = 0):
    """start(s), gpus, runtime(s) — arrival pattern like the reference
    trace (bursty arrivals, mixed job sizes)."""
    rng = random.Random(seed)
    t = 0.0
    out = []
    for _ in range(jobs):
        t += rng.expovariate(1 / 3.0)
        gpus = rng.choice([1, 1, 1, 1, 2, 4, 8])
        runtime = rng.uniform(30, 600)
        out.append((t, gpus, runtime))
    return out


def load_trace(path: str):
    out = []
    with open(path) as f:
        for line in f:
            parts = line.split()
            if len(parts) >= 3:
                out.append((float(parts[0]), int(float(parts[1])),
                            float(parts[2])))
    return out


def job_pods(rng, job_id: int, gpus: int):
    """Label sets for one trace job. Multi-GPU jobs are either one
    N-GPU pod or (half the time) a GANG of N whole-GPU pods — the
    reference's distributed e2e shape (test/distribute, one NCCL rank
    per GPU) — so the Permit barrier/expiry machinery is soaked too.
    Fractional jobs randomize request and priority like the reference
    simulator (simulator.py:64-69); some carry a lease_ms latency
    class."""
    prio = str(rng.choice([0, 50, 100]))
    if gpus > 1 and rng.random() < 0.5:
        gang = {C.POD_GPU_REQUEST: "1.0", C.POD_GPU_LIMIT: "1.0",
                C.POD_PRIORITY: prio,
                C.POD_GROUP_NAME: f"sim-gang-{job_id}",
                C.POD_GROUP_HEADCOUNT: str(gpus),
                C.POD_GROUP_THRESHOLD: "1.0"}
        return [dict(gang) for _ in range(gpus)]
    if gpus > 1:
        return [{C.POD_GPU_REQUEST: f"{gpus}.0",
                 C.POD_GPU_LIMIT: f"{gpus}.0", C.POD_PRIORITY: prio}]
    labels = {C.POD_GPU_REQUEST: str(rng.choice([0.25, 0.5, 0.75, 1.0])),
              C.POD_GPU_LIMIT: "1.0",
              C.POD_PRIORITY: str(rng.choice([0, 0, 100]))}
    if rng.random() < 0.2:
        labels[C.POD_LEASE_MS] = str(rng.choice([25, 50, 100]))
    return [labels]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trace", default="")
    ap.add_argument("--jobs", type=int, default=500)
    ap.add_argument("--nodes", type=int, default=4)
    ap.add_argument("--gpus-per-node", type=int, default=8)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--time-scale", type=float, default=0.0,
                    help="0 = as fast as possible (virtual time)")
    args = ap.parse_args()

    trace = load_trace(args.trace) if args.trace \
        else synthetic_trace(args.jobs, args.seed)
    rng = random.Random(args.seed)
    fc = FakeCluster(nodes={f"node-{i}": {"gpus": args.gpus_per_node}
                            for i in range(args.nodes)})

    runtime_of = {}        # pod key -> job runtime (s)
    active = []            # (end_time, key) of bound pods
    in_active = set()
    cycles = []
    requeues = 0
    ev_seen = 0

    def absorb_bindings():
        """Move newly-bound pods (harness event log) into the active
        set with their job runtime."""
        nonlocal ev_seen
        for e in fc.events[ev_seen:]:
            if e[0] == "bind" and e[1] not in in_active:
                active.append((fc.clock + runtime_of.get(e[1], 120.0),
                               e[1]))
                in_active.add(e[1])
        ev_seen = len(fc.events)

    def reap(now):
        for end, key in [a for a in active if a[0] <= now]:
            fc.delete_pod(key)
            active.remove((end, key))
            in_active.discard(key)

    for i, (start_t, gpus, runtime) in enumerate(trace):
        fc.clock = start_t
        reap(start_t)
        for k, labels in enumerate(job_pods(rng, i, gpus)):
            pod = fc.add_pod("sim", f"job{i}-{k}", labels)
            runtime_of[pod.key] = runtime
        t0 = time.perf_counter()
        fc.schedule_pending(rounds=1)
        cycles.append((time.perf_counter() - t0) * 1000)
        absorb_bindings()
        # Unschedulable (incl. Permit-expired gang members) requeues —
        # the real scheduler's backoff queue
        for pod in fc.pods.values():
            if pod.phase == "Unschedulable":
                requeues += 1
                pod.phase = "Pending"

    # drain: keep reaping finished jobs and rescheduling until quiet
    for _ in range(500):
        pending = [p for p in fc.pods.values()
                   if p.phase in ("Pending", "Waiting")]
        if not pending or not active:
            break
        active.sort()
        fc.clock = active[0][0]
        reap(fc.clock)
        fc.schedule_pending(rounds=1)
        absorb_bindings()
        for pod in fc.pods.values():
            if pod.phase == "Unschedulable":
                pod.phase = "Pending"

    ever_bound = {e[1] for e in fc.events if e[0] == "bind"}
    never = [p for p in fc.pods.values()
             if p.phase in ("Pending", "Waiting")]
    result = {
        "jobs": len(trace),
        "pods_submitted": len(runtime_of),
        "pods_ever_bound": len(ever_bound),
        "never_bound_after_drain": len(never),
        "requeues": requeues,
        "mean_cycle_ms": round(sum(cycles) / len(cycles), 3),
        "p99_cycle_ms": round(sorted(cycles)[int(len(cycles) * 0.99)], 3),
    }
    print(json.dumps(result))
    return 0


if __name__ == "__main__":
    sys.exit(main())
