"""Layer-by-layer comparison of the fused BN+ReLU path vs stock, to
localize composition errors that per-op tests miss."""
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from kubeshare_amd import ops  # noqa: E402
from kubeshare_amd.models.resnet import Bottleneck  # noqa: E402


def cos(a, b):
    return torch.nn.functional.cosine_similarity(
        a.float().flatten(), b.float().flatten(), dim=0).item()


def maxdiff(a, b):
    return (a.float() - b.float()).abs().max().item()


def run_block(fused, with_down, x, dy, seed=0):
    torch.manual_seed(seed)
    blk = Bottleneck(256, 64, stride=2 if with_down else 1).cuda().to(
        memory_format=torch.channels_last)
    blk.fused_ops = fused
    xi = x.clone().requires_grad_()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = blk(xi)
    y.backward(dy[: y.shape[0], :, : y.shape[2], : y.shape[3]].to(y.dtype))
    return blk, y, xi.grad


def main():
    torch.backends.cudnn.benchmark = True
    for with_down in (False, True):
        x = torch.randn(8, 256, 28, 28, device="cuda").contiguous(
            memory_format=torch.channels_last)
        dy = torch.randn(8, 256, 28, 28, device="cuda").contiguous(
            memory_format=torch.channels_last)
        b1, y1, gx1 = run_block(False, with_down, x, dy)
        b2, y2, gx2 = run_block(True, with_down, x, dy)
        print(f"--- bottleneck with_down={with_down}")
        print(f" y dtype {y1.dtype} vs {y2.dtype}; "
              f"fwd maxdiff {maxdiff(y1, y2):.4f} cos {cos(y1, y2):.5f}")
        print(f" dx cos {cos(gx1, gx2):.5f}")
        for name in ("conv1", "conv2", "conv3"):
            g1 = getattr(b1, name).weight.grad
            g2 = getattr(b2, name).weight.grad
            print(f" {name}.w grad cos {cos(g1, g2):.5f} "
                  f"maxdiff {maxdiff(g1, g2):.4f}")
        for name in ("bn1", "bn2", "bn3"):
            g1 = getattr(b1, name).weight.grad
            g2 = getattr(b2, name).weight.grad
            print(f" {name}.gamma grad cos {cos(g1, g2):.5f}")
            print(f" {name} running_mean maxdiff "
                  f"{maxdiff(getattr(b1, name).running_mean, getattr(b2, name).running_mean):.5f}")

    # deeper stack: 3 blocks
    torch.manual_seed(1)
    s1 = torch.nn.Sequential(*[Bottleneck(256, 64) for _ in range(3)])\
        .cuda().to(memory_format=torch.channels_last)
    torch.manual_seed(1)
    s2 = torch.nn.Sequential(*[Bottleneck(256, 64) for _ in range(3)])\
        .cuda().to(memory_format=torch.channels_last)
    s2.load_state_dict(s1.state_dict())
    for m in s2.modules():
        if hasattr(m, "fused_ops"):
            m.fused_ops = True
    x = torch.randn(8, 256, 28, 28, device="cuda").contiguous(
        memory_format=torch.channels_last)
    for tag, s in (("stock", s1), ("fused", s2)):
        xi = x.clone().requires_grad_()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = s(xi)
        y.float().pow(2).mean().backward()
        if tag == "stock":
            y1, g1 = y, [m.weight.grad.clone() for m in s1.modules()
                         if isinstance(m, torch.nn.Conv2d)]
        else:
            y2, g2 = y, [m.weight.grad.clone() for m in s2.modules()
                         if isinstance(m, torch.nn.Conv2d)]
    print("--- 3-block stack")
    print(f" fwd maxdiff {maxdiff(y1, y2):.4f}")
    for i, (a, b) in enumerate(zip(g1, g2)):
        print(f" conv[{i}] grad cos {cos(a, b):.5f}")


if __name__ == "__main__":
    main()
