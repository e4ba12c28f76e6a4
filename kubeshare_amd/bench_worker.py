"""bench_worker — one co-located "pod": trains a model on synthetic data
under the LD_PRELOAD isolation hook, paced by bench.py over stdin/stdout.

Protocol with the parent (bench.py / tests):
    worker -> parent: READY\n                (after warmup + sync)
    parent -> worker: GO\n
    worker -> parent: DONE <elapsed_s> <images> <loss>\n

Run exactly --steps full training steps (forward + loss + backward +
optimizer step) between GO and DONE; no work is skipped inside the
timed region.
"""
from __future__ import annotations

import argparse
import ctypes
import os
import sys
import time


def log(msg):
    print(f"[worker {os.environ.get('POD_NAME','?')}] {msg}",
          file=sys.stderr, flush=True)


def check_hook_active(required: bool):
    """On a GPU box the LD_PRELOAD hook must actually be attached —
    fail loudly rather than silently running un-isolated."""
    try:
        lib = ctypes.CDLL(None)
        fn = lib.ks_hook_active
        fn.restype = ctypes.c_int
        active = fn()
    except (OSError, AttributeError):
        active = -1
    if required and active != 1:
        raise RuntimeError(
            f"libhiphook not attached (ks_hook_active={active}) but this "
            f"worker was started with isolation required")
    return active


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50",
                    choices=["resnet18", "resnet50", "vgg16"],
                    help="ImageNet-shaped families only (the mnist/"
                         "lstm families have their own input shapes — "
                         "see models.small.synthetic_batch)")
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--image-size", type=int, default=224)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--no-channels-last", action="store_true")
    ap.add_argument("--lr", type=float, default=0.02)
    ap.add_argument("--use-ops", default="auto", choices=["auto", "on", "off"],
                    help="use the kubeshare_amd HIP fused ops")
    ap.add_argument("--fused-sgd", action="store_true",
                    help="use the HIP multi-tensor SGD instead of torch's")
    args = ap.parse_args()

    from kubeshare_amd.utils.tuning import apply_miopen_tuning
    apply_miopen_tuning()  # before the first conv

    import torch

    torch.backends.cudnn.benchmark = True  # MIOpen find+cache per shape
    dev = torch.device(args.device)
    on_gpu = dev.type == "cuda"
    if on_gpu:
        check_hook_active(os.environ.get("KUBESHARE_REQUIRE_HOOK") == "1")

    from kubeshare_amd.models import build_model
    # stable per-pod seed (hash() is salt-randomized per process):
    # per_pod_loss in the bench output is reproducible across runs
    import zlib
    torch.manual_seed(zlib.crc32(
        os.environ.get("POD_NAME", "").encode()) % 2**31)
    model = build_model(args.model)
    use_ops = False
    if on_gpu and args.use_ops != "off":
        try:
            from kubeshare_amd import ops
            model = ops.fuse_model(model)
            use_ops = True
        except Exception as e:  # noqa: BLE001
            if args.use_ops == "on":
                raise
            log(f"ops unavailable ({e}); stock modules")
    model = model.to(dev)
    channels_last = on_gpu and not args.no_channels_last
    if channels_last:
        model = model.to(memory_format=torch.channels_last)

    if use_ops and args.fused_sgd:
        from kubeshare_amd import ops as _ops
        opt = _ops.FusedSGD(model.parameters(), lr=args.lr, momentum=0.9,
                            weight_decay=1e-4)
    else:
        opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9,
                              weight_decay=1e-4)
    loss_fn = torch.nn.CrossEntropyLoss()

    # synthetic data, fixed on-device batch (BASELINE: no network for
    # datasets; data generation is outside the measured contract)
    x = torch.randn(args.batch, 3, args.image_size, args.image_size,
                    device=dev)
    if channels_last:
        x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (args.batch,), device=dev)

    amp_dtype = torch.bfloat16 if args.dtype == "bf16" else None

    def step():
        opt.zero_grad(set_to_none=True)
        if amp_dtype is not None:
            with torch.autocast(device_type=dev.type, dtype=amp_dtype):
                out = model(x)
                loss = loss_fn(out, y)
        else:
            out = model(x)
            loss = loss_fn(out, y)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        loss = step()
    if on_gpu:
        torch.cuda.synchronize()

    log(f"warmed up (ops={'on' if use_ops else 'off'}, "
        f"channels_last={channels_last})")
    print("READY", flush=True)
    line = sys.stdin.readline()
    if not line.startswith("GO"):
        log(f"unexpected command {line!r}; exiting")
        return 1

    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    if on_gpu:
        torch.cuda.synchronize()
    t1 = time.perf_counter()

    images = args.steps * args.batch
    print(f"DONE {t1 - t0:.6f} {images} {loss.item():.4f}", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
