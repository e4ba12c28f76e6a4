"""Sweep ResNet50 training-step configurations on one GPU (solo, no
isolation) to pick bench.py defaults: batch size, memory format, MIOpen
find mode, cudnn.benchmark. Prints one JSON line per config.

    gpurun -- 'python tools/model_speed.py --quick'
"""
import argparse
import itertools
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def run_cfg(model_name, batch, channels_last, benchmark, dtype, steps,
            warmup, graph=False, fused=False):
    from kubeshare_amd.utils.tuning import apply_miopen_tuning
    apply_miopen_tuning()
    import torch
    from kubeshare_amd.models import build_model

    torch.backends.cudnn.benchmark = benchmark
    dev = torch.device("cuda:0")
    model = build_model(model_name)
    if fused:
        from kubeshare_amd import ops
        model = ops.fuse_model(model)
    model = model.to(dev)
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=0.02, momentum=0.9)
    x = torch.randn(batch, 3, 224, 224, device=dev)
    if channels_last:
        x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), device=dev)
    amp = torch.bfloat16 if dtype == "bf16" else None

    def fwd_bwd():
        if amp:
            with torch.autocast("cuda", dtype=amp):
                loss = torch.nn.functional.cross_entropy(model(x), y)
        else:
            loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        return loss

    def step():
        opt.zero_grad(set_to_none=True)
        fwd_bwd()
        opt.step()

    g = None
    if graph:
        # hipGraph-captured whole step: zero_grad must keep buffers
        # (static addresses) and warmup must run on a side stream
        for _ in range(max(3, warmup)):
            step()
        opt.zero_grad(set_to_none=False)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                for p in model.parameters():
                    p.grad.zero_()
                fwd_bwd()
                opt.step()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        with torch.cuda.graph(g):
            for p in model.parameters():
                p.grad.zero_()
            fwd_bwd()
            opt.step()

        def step():  # noqa: F811
            g.replay()

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    del model, opt, x, y, g
    torch.cuda.empty_cache()
    return {
        "model": model_name, "batch": batch, "channels_last": channels_last,
        "benchmark": benchmark, "dtype": dtype, "graph": graph,
        "fused": fused,
        "ms_per_step": round(dt / steps * 1000, 2),
        "images_per_s": round(batch * steps / dt, 1),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--graphs", action="store_true",
                    help="compare eager vs hipGraph-captured step")
    ap.add_argument("--one", action="store_true",
                    help="single canonical config (for rocprof)")
    ap.add_argument("--fused", action="store_true",
                    help="use the kubeshare_amd fused BN+ReLU kernels")
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=6)
    args = ap.parse_args()

    if args.one:
        cfgs = [(args.batch, True, True, "bf16", False)]
    elif args.graphs:
        cfgs = [(args.batch, True, True, "bf16", False),
                (args.batch, True, True, "bf16", True),
                (512, True, True, "bf16", True)]
    elif args.quick:
        grid = [(256, True, True), (256, False, True), (256, True, False),
                (64, True, True), (512, True, True)]
        cfgs = [(b, cl, bm, "bf16", False) for b, cl, bm in grid]
    else:
        cfgs = [(b, cl, bm, d, False) for b, cl, bm, d in itertools.product(
            [64, 128, 256, 512], [True, False], [True, False], ["bf16"])]

    for batch, cl, bm, dtype, graph in cfgs:
        try:
            r = run_cfg(args.model, batch, cl, bm, dtype, args.steps,
                        args.warmup, graph=graph, fused=args.fused)
        except RuntimeError as e:
            r = {"batch": batch, "channels_last": cl, "benchmark": bm,
                 "graph": graph, "error": str(e)[:300]}
        print(json.dumps(r), flush=True)


if __name__ == "__main__":
    main()
