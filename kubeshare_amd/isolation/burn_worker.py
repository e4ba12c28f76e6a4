"""burn_worker — a minimal GPU-bound "pod" for isolation tests: loops
the calibrated ks_ops burn kernel under the LD_PRELOAD hook for a fixed
wall duration, then reports how much GPU time it got.

    python -m kubeshare_amd.isolation.burn_worker --duration-ms 5000
Prints: BURNED <wall_s> <iterations> <leases> <used_ms> <queued_ms>
(queued_ms = GPU time actually submitted — iters x burn_ms — the
client-side ground truth for lease-accounting accuracy tests).

--duty-cycle D (<1.0) makes the worker bursty: bursts of work followed
by CPU-idle sleeps so the submitted GPU time is ~D of wall — the
workload class where wall-charged leases over-report.
"""
import argparse
import ctypes
import sys
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration-ms", type=float, default=5000)
    ap.add_argument("--burn-ms", type=float, default=5.0)
    ap.add_argument("--duty-cycle", type=float, default=1.0)
    ap.add_argument("--burst-ms", type=float, default=90.0,
                    help="burst length when duty-cycle < 1")
    ap.add_argument("--wait-go", action="store_true",
                    help="print READY after warmup and wait for GO on stdin")
    args = ap.parse_args()

    import torch
    from kubeshare_amd import ops

    assert torch.cuda.is_available()
    torch.cuda.init()
    ops.burn(1.0)
    torch.cuda.synchronize()

    if args.wait_go:
        print("READY", flush=True)
        if not sys.stdin.readline().startswith("GO"):
            return 1

    t0 = time.perf_counter()
    iters = 0
    deadline = t0 + args.duration_ms / 1000.0
    while time.perf_counter() < deadline:
        if args.duty_cycle < 1.0:
            # one burst of ~burst_ms GPU work, then a CPU-idle gap so
            # submitted GPU time ~= duty_cycle x wall
            burst_start = time.perf_counter()
            n = max(1, int(args.burst_ms / args.burn_ms))
            for _ in range(n):
                ops.burn(args.burn_ms)
                iters += 1
            torch.cuda.synchronize()
            burst_wall = time.perf_counter() - burst_start
            time.sleep(burst_wall * (1.0 - args.duty_cycle)
                       / args.duty_cycle)
        else:
            ops.burn(args.burn_ms)
            iters += 1
            if iters % 8 == 0:
                torch.cuda.synchronize()  # bound queue-ahead
    torch.cuda.synchronize()
    t1 = time.perf_counter()

    lib = ctypes.CDLL(None)
    lib.ks_hook_leases.restype = ctypes.c_longlong
    lib.ks_hook_used_ms.restype = ctypes.c_double
    leases = lib.ks_hook_leases()
    used = lib.ks_hook_used_ms()
    queued = iters * args.burn_ms
    print(f"BURNED {t1 - t0:.3f} {iters} {leases} {used:.1f} {queued:.1f}",
          flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
