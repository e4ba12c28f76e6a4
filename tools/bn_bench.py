"""Per-shape microbench: fused ks_ops BN+ReLU vs the stock
MIOpen/elementwise sequence, forward and forward+backward, on ResNet50's
BN shapes at batch 256. Prints one JSON line per shape."""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

SHAPES = [  # (N, C, H, W) — ResNet50 @ bs256, one per stage
    (256, 64, 112, 112),
    (256, 256, 56, 56),
    (256, 128, 28, 28),
    (256, 512, 28, 28),
    (256, 1024, 14, 14),
    (256, 2048, 7, 7),
]


def bench(fn, sync, steps=20, warmup=5):
    for _ in range(warmup):
        fn()
    sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    sync()
    return (time.perf_counter() - t0) / steps * 1000


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    args = ap.parse_args()

    import torch
    from kubeshare_amd import ops

    dev = "cuda"
    sync = torch.cuda.synchronize
    for (n, c, h, w) in SHAPES:
        n = args.batch
        x = torch.randn(n, c, h, w, device=dev).to(torch.bfloat16)\
            .contiguous(memory_format=torch.channels_last)
        res = torch.randn_like(x)
        bn1 = torch.nn.BatchNorm2d(c).to(dev)
        bn2 = torch.nn.BatchNorm2d(c).to(dev)
        gb = torch.randn_like(x)

        def stock_fwd():
            with torch.autocast("cuda", dtype=torch.bfloat16):
                return torch.relu(bn1(x) + res)

        def fused_fwd():
            return ops.bn_relu(x, bn2, res=res)

        xg = x.clone().requires_grad_()
        resg = res.clone().requires_grad_()

        def stock_fb():
            with torch.autocast("cuda", dtype=torch.bfloat16):
                y = torch.relu(bn1(xg) + resg)
            y.backward(gb)
            xg.grad = None
            resg.grad = None

        def fused_fb():
            y = ops.bn_relu(xg, bn2, res=resg)
            y.backward(gb)
            xg.grad = None
            resg.grad = None

        # GPU-only throughput: launch L iterations back-to-back, time
        # with events (host launch gaps overlap, so this is kernel time)
        ext = ops._load()
        w32 = bn2.weight.detach().float()
        b32 = bn2.bias.detach().float()

        def gpu_only(fn, iters=30):
            fn()
            sync()
            ev0 = torch.cuda.Event(enable_timing=True)
            ev1 = torch.cuda.Event(enable_timing=True)
            ev0.record()
            for _ in range(iters):
                fn()
            ev1.record()
            sync()
            return ev0.elapsed_time(ev1) / iters

        def ext_fwd():
            ext.bn_relu_fwd_train(x, w32, b32, bn2.running_mean,
                                  bn2.running_var, 0.1, 1e-5, res, True)

        def aten_fwd():
            torch.ops.aten.miopen_batch_norm(
                x.float(), bn1.weight, bn1.bias, bn1.running_mean,
                bn1.running_var, True, 0.1, 1e-5)

        r = {
            "shape": [n, c, h, w],
            "mb": round(x.numel() * 2 / 1e6, 1),
            "stock_fwd_ms": round(bench(stock_fwd, sync), 3),
            "fused_fwd_ms": round(bench(fused_fwd, sync), 3),
            "stock_fb_ms": round(bench(stock_fb, sync), 3),
            "fused_fb_ms": round(bench(fused_fb, sync), 3),
            "gpuonly_ext_fwd_ms": round(gpu_only(ext_fwd), 3),
            "gpuonly_aten_fwd_ms": round(gpu_only(aten_fwd), 3),
        }
        r["fwd_speedup"] = round(r["stock_fwd_ms"] / r["fused_fwd_ms"], 2)
        r["fb_speedup"] = round(r["stock_fb_ms"] / r["fused_fb_ms"], 2)
        print(json.dumps(r), flush=True)
        del x, res, gb, xg, resg
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
