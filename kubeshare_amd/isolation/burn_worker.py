"""burn_worker — a minimal GPU-bound "pod" for isolation tests: loops
the calibrated ks_ops burn kernel under the LD_PRELOAD hook for a fixed
wall duration, then reports how much GPU time it got.

    python -m kubeshare_amd.isolation.burn_worker --duration-ms 5000
Prints: BURNED <wall_s> <iterations> <leases> <used_ms>
"""
import argparse
import ctypes
import sys
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration-ms", type=float, default=5000)
    ap.add_argument("--burn-ms", type=float, default=5.0)
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
    print(f"BURNED {t1 - t0:.3f} {iters} {leases} {used:.1f}", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
