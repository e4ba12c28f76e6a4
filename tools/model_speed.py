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
            warmup):
    import torch
    from kubeshare_amd.models import build_model

    torch.backends.cudnn.benchmark = benchmark
    dev = torch.device("cuda:0")
    model = build_model(model_name).to(dev)
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=0.02, momentum=0.9)
    x = torch.randn(batch, 3, 224, 224, device=dev)
    if channels_last:
        x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), device=dev)
    amp = torch.bfloat16 if dtype == "bf16" else None

    def step():
        opt.zero_grad(set_to_none=True)
        if amp:
            with torch.autocast("cuda", dtype=amp):
                loss = torch.nn.functional.cross_entropy(model(x), y)
        else:
            loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    del model, opt, x, y
    torch.cuda.empty_cache()
    return {
        "model": model_name, "batch": batch, "channels_last": channels_last,
        "benchmark": benchmark, "dtype": dtype,
        "ms_per_step": round(dt / steps * 1000, 2),
        "images_per_s": round(batch * steps / dt, 1),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=6)
    args = ap.parse_args()

    if args.quick:
        grid = [(256, True, True), (256, False, True), (256, True, False),
                (64, True, True), (512, True, True)]
        cfgs = [(b, cl, bm, "bf16") for b, cl, bm in grid]
    else:
        cfgs = [(b, cl, bm, d) for b, cl, bm, d in itertools.product(
            [64, 128, 256, 512], [True, False], [True, False], ["bf16"])]

    for batch, cl, bm, dtype in cfgs:
        try:
            r = run_cfg(args.model, batch, cl, bm, dtype, args.steps,
                        args.warmup)
        except RuntimeError as e:
            r = {"batch": batch, "channels_last": cl, "benchmark": bm,
                 "error": str(e)[:200]}
        print(json.dumps(r), flush=True)


if __name__ == "__main__":
    main()
