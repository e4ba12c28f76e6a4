"""DDP-gang-under-isolation probe (round-2 validation tool).

Launches a torchrun DDP job (RCCL) where each rank runs under the
LD_PRELOAD hook, either one rank per GPU (the supported gang config) or
two ranks sharing one GPU (the hazardous config documented in
docs/ROADMAP.md: live via gpu-schd's revocation, but slow). Reports
per-step time so the penalty is measurable.

    # on an 8-GPU box:
    python tools/ddp_gang.py --ranks 2 --share-gpu        # hazard probe
    python tools/ddp_gang.py --ranks 2                    # 1 rank/GPU
"""
import argparse
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def worker():
    import torch
    import torch.distributed as dist

    from kubeshare_amd.models import resnet18

    dist.init_process_group("nccl")
    torch.cuda.set_device(0)  # ROCR_VISIBLE_DEVICES narrows the view
    model = torch.nn.parallel.DistributedDataParallel(
        resnet18().cuda().to(memory_format=torch.channels_last))
    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    x = torch.randn(32, 3, 224, 224, device="cuda").contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (32,), device="cuda")
    steps = int(os.environ.get("DDP_STEPS", "10"))
    for _ in range(3):
        _step(model, opt, x, y)
    torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(steps):
        _step(model, opt, x, y)
    torch.cuda.synchronize()
    dist.barrier()
    dt = (time.perf_counter() - t0) / steps * 1000
    if dist.get_rank() == 0:
        print(f"DDP_RESULT ms_per_step={dt:.2f}", flush=True)
    dist.destroy_process_group()


def _step(model, opt, x, y):
    import torch
    opt.zero_grad(set_to_none=True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = torch.nn.functional.cross_entropy(model(x), y)
    loss.backward()
    opt.step()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ranks", type=int, default=2)
    ap.add_argument("--share-gpu", action="store_true",
                    help="all ranks on GPU 0 (hazard probe)")
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--timeout", type=float, default=300)
    args = ap.parse_args()

    from kubeshare_amd.isolation.local import LocalGPUShare

    shares = []
    procs = []
    try:
        if args.share_gpu:
            share = LocalGPUShare(gpu_index=0).start()
            shares.append(share)
        else:
            for i in range(args.ranks):
                shares.append(LocalGPUShare(gpu_index=i).start())
        for r in range(args.ranks):
            share = shares[0] if args.share_gpu else shares[r]
            gpu = 0 if args.share_gpu else r
            h = share.add_pod(f"gang/rank{r}", request=1.0 / args.ranks
                              if args.share_gpu else 1.0, limit=1.0)
            env = h.env(gpu_index=gpu)
            env.update({
                "RANK": str(r), "LOCAL_RANK": str(r),
                "WORLD_SIZE": str(args.ranks),
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29571",
                "DDP_STEPS": str(args.steps),
                "PYTHONPATH": REPO + os.pathsep + env.get("PYTHONPATH", ""),
            })
            procs.append(subprocess.Popen(
                [sys.executable, __file__, "--worker"], env=env, cwd=REPO))
        deadline = time.time() + args.timeout
        for p in procs:
            p.wait(timeout=max(1.0, deadline - time.time()))
        print("GANG_OK" if all(p.returncode == 0 for p in procs)
              else "GANG_FAILED", flush=True)
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
        for s in shares:
            s.stop()


if __name__ == "__main__":
    if "--worker" in sys.argv:
        worker()
    else:
        main()
