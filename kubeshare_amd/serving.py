"""Serving under fractional sharing — the latency side of the stack.

A serving pod is the bursty workload class: requests arrive
intermittently, each needs the GPU for a few ms, and what matters is
tail latency while a co-located training pod is free to soak the rest
of the GPU. Three pieces of the isolation design exist for exactly
this case:
  - idle release: the trainer's lease is reclaimable ~25 ms after its
    dispatch stream pauses, and the server pod's own lease returns as
    soon as it goes quiet — neither sits on the token;
  - hint-sized leases: the server pod's REQ carries its per-lease busy
    EWMA (a few ms), so gpu-schd grants it short leases instead of
    300 ms defaults;
  - hipGraph: the model forward is captured once (capture bypasses the
    gate: no GPU time) and each request replays it as ONE gated
    dispatch (hipGraphLaunch) — launch overhead off the hot path.

The remaining latency floor is the scheduler's base quota -q: a
request that arrives while the trainer holds its lease waits for the
trainer's drain, so p99 ~ trainer lease length. `serve_probe`
measures that tradeoff (BASELINE.md serving table): -q 300 ms is the
throughput end, -q 25-50 ms the latency end.
"""
from __future__ import annotations

import time


class GraphReplayServer:
    """Capture a model's forward in a hipGraph once; serve() replays it
    (one gated dispatch per request)."""

    def __init__(self, model, example_input, warmup: int = 3):
        import torch
        self.torch = torch
        self.model = model.eval()
        self.static_in = example_input.clone()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            with torch.no_grad():
                for _ in range(warmup):
                    self.model(self.static_in)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.no_grad():
            with torch.cuda.graph(self.graph):
                self.static_out = self.model(self.static_in)

    def serve(self, x) -> "object":
        """One request: copy in, replay, sync. Returns the STATIC
        output buffer — the next request overwrites it; callers that
        keep results across requests must .clone()."""
        self.static_in.copy_(x)
        self.graph.replay()
        self.torch.cuda.synchronize()
        return self.static_out

    def bench(self, x, n_requests: int = 100,
              interarrival_s: float = 0.05) -> dict:
        import ctypes
        try:
            lib = ctypes.CDLL(None)
            lib.ks_hook_wait_ms.restype = ctypes.c_double
            wait_ms = lib.ks_hook_wait_ms
        except (OSError, AttributeError):
            wait_ms = lambda: 0.0  # noqa: E731 — hook not attached
        lat, waits = [], []
        for _ in range(n_requests):
            w0 = wait_ms()
            t0 = time.perf_counter()
            self.serve(x)
            lat.append((time.perf_counter() - t0) * 1000.0)
            waits.append(wait_ms() - w0)
            time.sleep(interarrival_s)
        outliers = [(i, round(v, 1), round(waits[i], 1))
                    for i, v in enumerate(lat) if v > 200.0]
        lat.sort()

        def pct(p):
            return lat[min(len(lat) - 1, int(p / 100.0 * len(lat)))]
        return {"n": len(lat), "p50_ms": round(pct(50), 2),
                "p95_ms": round(pct(95), 2), "p99_ms": round(pct(99), 2),
                "max_ms": round(lat[-1], 2),
                "mean_ms": round(sum(lat) / len(lat), 2),
                "outliers": outliers[:10]}
