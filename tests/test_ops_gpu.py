"""Numerics tests for the hand-written gfx950 kernels vs plain PyTorch
fp32 references (run on MI355X via gpurun)."""
import os
import sys

import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from kubeshare_amd import ops  # noqa: E402

SHAPES = [(8, 64, 56, 56), (4, 256, 28, 28), (2, 2048, 7, 7),
          (3, 8, 10, 10),
          # CG (=C/8) NOT dividing the 256-thread block: exercises the
          # inactive-thread guard (was a double-count bug)
          (4, 24, 17, 17), (2, 1536, 9, 9)]


def _mk(shape, seed=0):
    torch.manual_seed(seed)
    x = torch.randn(shape, device="cuda").to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    return x


def _ref_bn_relu(x32, bn_ref, res32=None, training=True):
    y = torch.nn.functional.batch_norm(
        x32, bn_ref["rm"], bn_ref["rv"], bn_ref["w"], bn_ref["b"],
        training, 0.1, 1e-5)
    if res32 is not None:
        y = y + res32
    return torch.relu(y)


@pytest.mark.parametrize("shape", SHAPES)
@pytest.mark.parametrize("with_res", [False, True])
def test_bn_relu_forward_backward(shape, with_res):
    C = shape[1]
    x = _mk(shape).requires_grad_()
    res = _mk(shape, seed=1).requires_grad_() if with_res else None
    bn = torch.nn.BatchNorm2d(C).cuda()
    with torch.no_grad():
        bn.weight.uniform_(0.5, 1.5)
        bn.bias.uniform_(-0.5, 0.5)

    # fp32 reference on the SAME bf16 values
    x32 = x.detach().float().requires_grad_()
    res32 = res.detach().float().requires_grad_() if with_res else None
    ref_state = {"w": bn.weight.detach().clone(),
                 "b": bn.bias.detach().clone(),
                 "rm": bn.running_mean.clone(), "rv": bn.running_var.clone()}
    ref_state["w"].requires_grad_()
    ref_state["b"].requires_grad_()
    y_ref = _ref_bn_relu(x32, ref_state, res32)
    # identical (bf16-rounded) upstream gradient for both paths
    gb = torch.randn_like(y_ref).to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    y_ref.backward(gb.float())

    y = ops.bn_relu(x, bn, res=res)
    y.backward(gb)
    torch.cuda.synchronize()

    assert y.dtype == torch.bfloat16
    torch.testing.assert_close(y.float(), y_ref, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(bn.running_mean, ref_state["rm"],
                               rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(bn.running_var, ref_state["rv"],
                               rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(x.grad.float(), x32.grad,
                               rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(bn.weight.grad, ref_state["w"].grad,
                               rtol=2e-2, atol=2e-1)
    torch.testing.assert_close(bn.bias.grad, ref_state["b"].grad,
                               rtol=2e-2, atol=2e-1)
    if with_res:
        torch.testing.assert_close(res.grad.float(), res32.grad,
                                   rtol=5e-2, atol=5e-2)


def test_sgd_momentum_channels_last():
    """FusedSGD on channels-last conv weights (the flagship layout):
    matches torch.optim.SGD."""
    torch.manual_seed(3)
    conv = torch.nn.Conv2d(64, 128, 3, bias=False).cuda().to(
        memory_format=torch.channels_last)
    ref = conv.weight.detach().clone()
    g = torch.randn_like(conv.weight)

    ref_p = ref.clone().requires_grad_()
    ref_p.grad = g.clone()
    opt = torch.optim.SGD([ref_p], lr=0.1, momentum=0.9, weight_decay=1e-4)
    for _ in range(3):
        opt.step()

    fused = ops.FusedSGD([conv.weight.requires_grad_()], lr=0.1,
                         momentum=0.9, weight_decay=1e-4)
    conv.weight.grad = g.clone()
    for _ in range(3):
        fused.step()
    torch.cuda.synchronize()
    torch.testing.assert_close(conv.weight, ref_p, rtol=1e-5, atol=1e-6)


def test_bn_relu_eval_mode():
    x = _mk((4, 64, 14, 14))
    bn = torch.nn.BatchNorm2d(64).cuda()
    with torch.no_grad():
        bn.running_mean.uniform_(-1, 1)
        bn.running_var.uniform_(0.5, 2.0)
        bn.weight.uniform_(0.5, 1.5)
    bn.eval()
    ref_state = {"w": bn.weight.detach(), "b": bn.bias.detach(),
                 "rm": bn.running_mean.clone(),
                 "rv": bn.running_var.clone()}
    y_ref = _ref_bn_relu(x.float(), ref_state, training=False)
    with torch.no_grad():
        y = ops.bn_relu(x, bn)
    torch.testing.assert_close(y.float(), y_ref, rtol=2e-2, atol=2e-2)


def _cos(a, b):
    return torch.nn.functional.cosine_similarity(
        a.float().flatten(), b.float().flatten(), dim=0).item()


@pytest.mark.parametrize("with_down", [False, True])
def test_fused_bottleneck_composes(with_down):
    """One full Bottleneck block fused vs stock under bf16 autocast:
    forward, input grad and every parameter grad must agree to bf16
    tolerance. (Whole-50-layer grad comparison is NOT meaningful: the
    stock autocast path keeps BN internals in fp32 while ours rounds
    y/dym/dx to bf16 once per layer, and the per-layer ~1e-3 cosine loss
    compounds over depth — measured decay in tools/debug_fused.py.)"""
    from kubeshare_amd.models.resnet import Bottleneck

    outs = []
    for fused in (False, True):
        torch.manual_seed(0)
        blk = Bottleneck(256, 64, stride=2 if with_down else 1).cuda().to(
            memory_format=torch.channels_last)
        blk.fused_ops = fused
        torch.manual_seed(7)
        x = torch.randn(8, 256, 28, 28, device="cuda").contiguous(
            memory_format=torch.channels_last).requires_grad_()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = blk(x)
        dy = torch.ones_like(y)
        y.backward(dy)
        outs.append((y.float(), x.grad,
                     {n: p.grad.clone() for n, p in blk.named_parameters()}))
    (y1, gx1, g1), (y2, gx2, g2) = outs
    assert _cos(y1, y2) > 0.999
    assert _cos(gx1, gx2) > 0.99
    for name in g1:
        assert _cos(g1[name], g2[name]) > 0.98, name


def test_fused_resnet50_trains():
    """Trainability: 5 fused training steps must keep reducing the loss
    on a fixed batch, and match the stock model's loss trajectory to
    bf16 tolerance."""
    from kubeshare_amd.models import resnet50
    torch.manual_seed(0)
    m1 = resnet50(num_classes=100).cuda().to(
        memory_format=torch.channels_last)
    m2 = resnet50(num_classes=100).cuda().to(
        memory_format=torch.channels_last)
    m2.load_state_dict(m1.state_dict())
    ops.fuse_model(m2)
    x = torch.randn(16, 3, 128, 128, device="cuda").contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 100, (16,), device="cuda")

    traj = []
    for m in (m1, m2):
        opt = torch.optim.SGD(m.parameters(), lr=0.01, momentum=0.9)
        losses = []
        for _ in range(6):
            opt.zero_grad(set_to_none=True)
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = torch.nn.functional.cross_entropy(m(x), y)
            loss.backward()
            opt.step()
            losses.append(loss.item())
        traj.append(losses)
    torch.cuda.synchronize()
    stock, fused = traj
    assert fused[-1] < fused[0], f"fused loss not decreasing: {fused}"
    assert stock[-1] < stock[0], f"stock loss not decreasing: {stock}"
    # early-step agreement (before bf16 noise compounds through the
    # parameter trajectory)
    for a, b in list(zip(stock, fused))[:3]:
        assert abs(a - b) < 0.3, (stock, fused)


def test_bn_relu_odd_channels_falls_back():
    """C not divisible by 8: eager fallback, still correct."""
    x = torch.randn(2, 12, 9, 9, device="cuda").to(
        torch.bfloat16).contiguous(memory_format=torch.channels_last)
    bn = torch.nn.BatchNorm2d(12).cuda()
    y = ops.bn_relu(x, bn)
    assert y.shape == x.shape
    assert (y.float() >= 0).all()


def test_bn_only_no_relu_matches_torch():
    """relu=False variant (downsample-path BN, no activation) vs plain
    fp32 torch reference — forward and all gradients."""
    shape = (16, 64, 28, 28)
    x = _mk(shape).requires_grad_()
    bn = torch.nn.BatchNorm2d(shape[1]).cuda()
    with torch.no_grad():
        bn.weight.uniform_(0.5, 1.5)
        bn.bias.uniform_(-0.5, 0.5)
    x32 = x.detach().float().requires_grad_()
    w32 = bn.weight.detach().clone().requires_grad_()
    b32 = bn.bias.detach().clone().requires_grad_()
    rm, rv = bn.running_mean.clone(), bn.running_var.clone()
    y_ref = torch.nn.functional.batch_norm(x32, rm, rv, w32, b32,
                                           True, 0.1, 1e-5)
    gb = torch.randn_like(y_ref).to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    y_ref.backward(gb.float())

    y = ops.bn_relu(x, bn, relu=False)
    y.backward(gb)
    torch.cuda.synchronize()
    assert (y.float() < 0).any(), "no-relu output must keep negatives"
    torch.testing.assert_close(y.float(), y_ref, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(x.grad.float(), x32.grad,
                               rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(bn.weight.grad.float(), w32.grad,
                               rtol=2e-2, atol=2e-1)
    torch.testing.assert_close(bn.bias.grad.float(), b32.grad,
                               rtol=2e-2, atol=2e-1)
    torch.testing.assert_close(bn.running_mean, rm, rtol=1e-3, atol=1e-3)
