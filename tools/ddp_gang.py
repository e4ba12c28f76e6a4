"""DDP-gang-under-isolation probe (CLI over kubeshare_amd.parallel).

Launches a DDP job (RCCL over xGMI) where each rank runs under the
LD_PRELOAD hook, either one rank per GPU (the supported gang config) or
N ranks sharing ONE GPU (the risky config: SURVEY.md §2.4(b) — gated
collectives would deadlock; survives via the hook's librccl exemption
and gpu-schd's gang co-granting). Prints per-step time and the
server-side quota stats.

    python tools/ddp_gang.py --ranks 2 --share-gpu   # sharing probe
    python tools/ddp_gang.py --ranks 2               # 1 rank/GPU
"""
import argparse
import json
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ranks", type=int, default=2)
    ap.add_argument("--share-gpu", action="store_true",
                    help="all ranks on GPU 0 (hazard probe)")
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--model", default="resnet18")
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--timeout", type=float, default=300)
    args = ap.parse_args()

    from kubeshare_amd.parallel import launch_gang
    ok, stats = launch_gang(ranks=args.ranks, share_gpu=args.share_gpu,
                            steps=args.steps, model=args.model,
                            batch=args.batch, timeout=args.timeout)
    for st in stats:
        print("SCHD_STATS " + json.dumps(st), flush=True)
    print("GANG_OK" if ok else "GANG_FAILED", flush=True)
    return 0 if ok else 1


if __name__ == "__main__":
    main()
