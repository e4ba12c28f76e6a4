"""DDP gang under the isolation chain.

`ddp_worker` is the per-rank training loop (RCCL on GPU, gloo on CPU);
`launch_gang` stands up the full sharing stack — gpu-schd, pod-mgr per
rank, LD_PRELOAD hook — and runs N ranks either one-per-GPU (the
supported gang config) or all on ONE GPU (the hazardous config:
two ranks sharing a device, where gating a collective would deadlock;
kept alive by the RCCL exemption + gang co-granting + the `group`
field of the per-UUID config file).

Used by tools/ddp_gang.py (CLI) and tests/test_gpu_isolation.py (the
RCCL-under-sharing hardware proof).
"""
from __future__ import annotations

import os
import subprocess
import sys
import time

_REPO = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def _step(model, opt, x, y, autocast):
    import torch
    opt.zero_grad(set_to_none=True)
    if autocast:
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
    else:
        loss = torch.nn.functional.cross_entropy(model(x), y)
    loss.backward()
    opt.step()
    return loss


def ddp_worker(model_name: str = "resnet18", batch: int = 32,
               image_size: int = 224, steps: int = 10,
               warmup: int = 3) -> float:
    """One DDP rank: init from the torchrun/env contract, train
    `steps` timed steps, return ms/step (rank 0 also prints a
    DDP_RESULT line)."""
    import torch
    import torch.distributed as dist

    from ..models import build_model

    on_gpu = torch.cuda.is_available()
    # RCCL refuses two ranks on one device ("Duplicate GPU detected" —
    # same restriction as NCCL; the reference's own gang workloads only
    # ever put one NCCL rank per GPU, test/distribute/*). For the
    # shared-GPU gang config the launcher therefore selects gloo
    # gradients over GPU compute; whole-GPU gangs use RCCL.
    backend = os.environ.get("KUBESHARE_DDP_BACKEND") or \
        ("nccl" if on_gpu else "gloo")
    dist.init_process_group(backend)
    dev = "cuda" if on_gpu else "cpu"
    if on_gpu:
        torch.cuda.set_device(0)  # ROCR_VISIBLE_DEVICES narrows the view
    model = build_model(model_name).to(dev)
    if on_gpu:
        model = model.to(memory_format=torch.channels_last)
    model = torch.nn.parallel.DistributedDataParallel(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    x = torch.randn(batch, 3, image_size, image_size, device=dev)
    if on_gpu:
        x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), device=dev)
    for _ in range(warmup):
        _step(model, opt, x, y, autocast=on_gpu)
    if on_gpu:
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(steps):
        _step(model, opt, x, y, autocast=on_gpu)
    if on_gpu:
        torch.cuda.synchronize()
    dist.barrier()
    dt = (time.perf_counter() - t0) / steps * 1000
    if dist.get_rank() == 0:
        print(f"DDP_RESULT ms_per_step={dt:.2f}", flush=True)
    dist.destroy_process_group()
    return dt


def launch_gang(ranks: int = 2, share_gpu: bool = False, steps: int = 10,
                model: str = "resnet18", batch: int = 32,
                image_size: int = 224, timeout: float = 300.0,
                master_port: int = 29571):
    """Run a DDP gang under the full isolation chain. Returns
    (all_ok, per_gpu_stats) where per_gpu_stats is each
    LocalGPUShare's final gpu-schd STATS dict (server-side quota
    accounting, keyed by pod)."""
    from ..isolation.local import LocalGPUShare

    shares, procs = [], []
    stats = []
    try:
        if share_gpu:
            shares.append(LocalGPUShare(gpu_index=0).start())
        else:
            for i in range(ranks):
                shares.append(LocalGPUShare(gpu_index=i).start())
        handles = []
        for r in range(ranks):
            share = shares[0] if share_gpu else shares[r]
            handles.append(share.add_pod(
                f"gang/rank{r}",
                request=(1.0 / ranks) if share_gpu else 1.0,
                limit=1.0))
        # the `group` field in the per-UUID config makes gpu-schd
        # co-grant the gang when ranks share a GPU — set BEFORE any
        # worker can REQ
        for share in shares:
            _mark_gang(share)
        time.sleep(0.2)  # let inotify reload land
        for r in range(ranks):
            gpu = 0 if share_gpu else r
            env = handles[r].env(gpu_index=gpu)
            env.update({
                "RANK": str(r), "LOCAL_RANK": str(r),
                "WORLD_SIZE": str(ranks),
                "MASTER_ADDR": "127.0.0.1",
                "MASTER_PORT": str(master_port),
                "PYTHONPATH": _REPO + os.pathsep +
                env.get("PYTHONPATH", ""),
            })
            if share_gpu:
                env["KUBESHARE_DDP_BACKEND"] = "gloo"  # see ddp_worker
            procs.append(subprocess.Popen(
                [sys.executable, "-c",
                 "from kubeshare_amd.parallel.ddp import ddp_worker; "
                 f"ddp_worker({model!r}, {batch}, {image_size}, {steps})"],
                env=env, cwd=_REPO))
        deadline = time.time() + timeout
        for p in procs:
            p.wait(timeout=max(1.0, deadline - time.time()))
        for share in shares:
            stats.append(share.stats())
        return all(p.returncode == 0 for p in procs), stats
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
        for s in shares:
            s.stop()


def _mark_gang(share):
    """Rewrite the share's per-UUID config with the gang group set
    (config-file 5th field, native/common/protocol.hpp)."""
    from ..configdaemon import files as F
    quotas = [F.PodQuota(h.name, h.limit, h.request, h.memory, group="gang")
              for h in share.pods.values()]
    F.write_gpu_config(share.config_dir, share.uuid, quotas)
