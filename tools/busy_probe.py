"""Probe the rocm_smi busy counters against a known GPU load — the
calibration evidence behind gpu-schd's lease accounting
(native/schd/busy_sampler.hpp).

Runs three 2-second phases — idle, 100% busy (ks_ops.burn), ~30% duty
bursts — and for each prints the instantaneous busy_percent samples and
the coarse-grain accumulated counter delta (expected: delta/100 =
busy_ms; rocm_smi.h: "every millisecond the firmware calculates % busy
and accumulates it").

    gpurun -- 'python tools/busy_probe.py > gpurun_out/busy_probe.txt'
"""
import ctypes
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


class UtilCounter(ctypes.Structure):
    _fields_ = [("type", ctypes.c_int32), ("pad", ctypes.c_int32),
                ("value", ctypes.c_uint64)]


def main():
    lib = ctypes.CDLL("librocm_smi64.so")
    assert lib.rsmi_init(0) == 0
    dev = 0

    def read_acc():
        c = UtilCounter(0, 0, 0)
        ts = ctypes.c_uint64(0)
        rc = lib.rsmi_utilization_count_get(dev, ctypes.byref(c), 1,
                                            ctypes.byref(ts))
        return rc, c.value, ts.value

    def read_pct():
        p = ctypes.c_uint32(0)
        rc = lib.rsmi_dev_busy_percent_get(dev, ctypes.byref(p))
        return rc, p.value

    import torch  # noqa: F401
    from kubeshare_amd import ops

    ops.burn(5.0)
    import torch as t
    t.cuda.synchronize()

    def phase(name, fn, seconds=2.0):
        rc0, a0, ts0 = read_acc()
        t0 = time.perf_counter()
        pcts = []
        end = t0 + seconds
        while time.perf_counter() < end:
            fn()
            _, p = read_pct()
            pcts.append(p)
        t.cuda.synchronize()
        rc1, a1, ts1 = read_acc()
        wall = (time.perf_counter() - t0) * 1000
        print(f"{name}: wall={wall:.0f}ms acc_rc={rc0},{rc1} "
              f"acc_delta={a1 - a0} acc_busy_ms={(a1 - a0) / 100.0:.0f} "
              f"ts_delta_ms={(ts1 - ts0) / 1e6:.0f} "
              f"pct_samples={pcts[:10]}... mean_pct="
              f"{sum(pcts) / max(1, len(pcts)):.0f}", flush=True)

    phase("idle", lambda: time.sleep(0.05))

    def busy():
        ops.burn(20.0)
        t.cuda.synchronize()
    phase("busy100", busy)

    def bursty():
        ops.burn(30.0)
        t.cuda.synchronize()
        time.sleep(0.070)
    phase("bursty30", bursty)


if __name__ == "__main__":
    main()
