"""Kernel-level breakdown of the ResNet50 training step via
torch.profiler (ROCm backend) — where do the ~40 ms go?

    gpurun -- 'python tools/torch_profile.py > gpurun_out/kernels.txt'
"""
import argparse
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--fused", action="store_true")
    args = ap.parse_args()

    from kubeshare_amd.utils.tuning import apply_miopen_tuning
    apply_miopen_tuning()
    import torch
    from torch.profiler import ProfilerActivity, profile

    from kubeshare_amd.models import build_model

    torch.backends.cudnn.benchmark = True
    dev = torch.device("cuda:0")
    model = build_model(args.model)
    if args.fused:
        from kubeshare_amd import ops
        model = ops.fuse_model(model)
    model = model.to(dev).to(memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=0.02, momentum=0.9)
    x = torch.randn(args.batch, 3, 224, 224, device=dev).contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (args.batch,), device=dev)

    def step():
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()

    for _ in range(8):
        step()
    torch.cuda.synchronize()

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=False) as prof:
        for _ in range(args.steps):
            step()
        torch.cuda.synchronize()

    print(prof.key_averages().table(sort_by="self_cuda_time_total",
                                    row_limit=40))


if __name__ == "__main__":
    main()
