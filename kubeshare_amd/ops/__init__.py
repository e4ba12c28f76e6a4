"""kubeshare_amd.ops — hand-written HIP/gfx950 kernels.

Build: in-tree (the built .so travels with the repo snapshot), via
torch.utils.cpp_extension with PYTORCH_ROCM_ARCH=gfx950. On a GPU box a
missing extension is a hard error (fail-loud policy): GPU
paths must never silently fall back to eager PyTorch.
"""
from __future__ import annotations

import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_BUILD_DIR = os.path.join(_HERE, "_build")
_SRCS = [os.path.join(_HERE, "hip", "ks_ops.hip"),
         os.path.join(_HERE, "hip", "bn_relu.hip")]

_ext = None


def build_extension(verbose: bool = False):
    """Compile the extension for gfx950 (works on a CPU-only box: hipcc
    cross-compiles). Called by __graft_entry__.build()."""
    import torch  # noqa: F401  (sets up the ROCm toolchain env)
    from torch.utils import cpp_extension

    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(_BUILD_DIR, exist_ok=True)
    return cpp_extension.load(
        name="ks_ops",
        sources=_SRCS,
        build_directory=_BUILD_DIR,
        extra_cflags=["-O3"],
        verbose=verbose,
        is_python_module=True,
    )


def _load():
    global _ext
    if _ext is not None:
        return _ext
    import torch

    so = os.path.join(_BUILD_DIR, "ks_ops.so")
    if os.path.exists(so):
        import importlib.util

        spec = importlib.util.spec_from_file_location("ks_ops", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
        return _ext
    if torch.cuda.is_available():
        # GPU box without a prebuilt .so: build now — never fall back
        # silently to eager PyTorch on the GPU path.
        _ext = build_extension()
        return _ext
    raise ImportError(
        "kubeshare_amd.ops extension not built; run __graft_entry__.build()")


def burn(ms: float, blocks: int = 1024, threads: int = 256):
    """Occupy the current GPU for ~ms milliseconds (calibrated load
    generator used by the isolation tests and rocprof quota proofs)."""
    _load().burn(float(ms), blocks, threads)


class FusedSGD:
    """Multi-tensor fused SGD+momentum (f32 master params).

    Matches torch.optim.SGD(momentum=mu, weight_decay=wd, dampening=0,
    nesterov=False) semantics; one kernel per tensor, no Python-side
    per-parameter loop allocations after the first step.

    graph_mode="auto" (default on CUDA): once the grad set is stable,
    the ~161 per-tensor update launches and the ~161 grad-zero memsets
    are captured into two hipGraphs and replayed as ONE gated dispatch
    each per step. Requires stable grad storage, so zero_grad() zeroes
    in place instead of dropping the tensors (backward then accumulates
    into the same buffers — pointer-stable, capture-safe).
    """

    def __init__(self, params, lr: float, momentum: float = 0.9,
                 weight_decay: float = 0.0, graph_mode: str = "auto"):
        import torch

        self.params = [p for p in params if p.requires_grad]
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        # momentum buffers share the parameter's layout (channels-last
        # conv weights included): the kernel updates flat dense storage
        self.momenta = [torch.zeros_like(p) for p in self.params]
        self._first = True
        self._graph_mode = graph_mode
        self._step_graph = None
        self._zero_graph = None
        self._graph_ptrs = None  # grad data_ptrs the graphs were baked on

    def _graphable(self):
        import torch
        if self._graph_mode == "off" or not torch.cuda.is_available():
            return False
        import os
        if os.environ.get("KUBESHARE_SGD_GRAPH", "1") == "0":
            return False
        return all(p.grad is not None and p.grad.is_cuda
                   for p in self.params)

    def zero_grad(self, set_to_none: bool = True):
        if self._zero_graph is not None and self._graph_ptrs == tuple(
                p.grad.data_ptr() if p.grad is not None else 0
                for p in self.params):
            self._zero_graph.replay()
            return
        for p in self.params:
            if p.grad is not None:
                if set_to_none and self._step_graph is None:
                    p.grad = None
                else:
                    # graph mode needs pointer-stable grads
                    p.grad.zero_()

    def _capture(self):
        import torch
        ext = _load()
        ps = [p.data for p in self.params]
        gs = [p.grad for p in self.params]
        vs = list(self.momenta)
        ptrs = tuple(g.data_ptr() for g in gs)
        # (the update kernel is already warm: capture happens right
        # after a full eager step)
        step_g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(step_g):
            ext.sgd_momentum_(ps, gs, vs, self.lr, self.momentum,
                              self.weight_decay)
        zero_g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(zero_g):
            for g in gs:
                g.zero_()
        self._step_graph = step_g
        self._zero_graph = zero_g
        self._graph_ptrs = ptrs

    def step(self):
        ext = _load()
        if self._step_graph is not None and self._graph_ptrs == tuple(
                p.grad.data_ptr() if p.grad is not None else 0
                for p in self.params):
            self._step_graph.replay()
            return
        ps, gs, vs = [], [], []
        for p, v in zip(self.params, self.momenta):
            if p.grad is None:
                continue
            ps.append(p.data)
            gs.append(p.grad)
            vs.append(v)
        if self._first:
            # torch SGD's first step sets v = g + wd*p (no mu*v term
            # since v starts at 0) — identical here because v==0.
            self._first = False
        ext.sgd_momentum_(ps, gs, vs, self.lr, self.momentum,
                          self.weight_decay)
        # after a full-set eager step with stable grads, capture the
        # graphs for subsequent steps
        if self._step_graph is None and self._graph_mode == "auto" and \
                len(gs) == len(self.params) and self._graphable():
            try:
                self._capture()
            except Exception:  # noqa: BLE001 — capture is an optimization
                self._graph_mode = "off"


# ---------------------------------------------------- fused BN+ReLU(+add)
def _bn_relu_autograd():
    """Lazily build the autograd.Function (torch import deferred)."""
    global _BNReLUFn
    if _BNReLUFn is not None:
        return _BNReLUFn
    import torch

    class BNReLUFn(torch.autograd.Function):
        """y = relu(batch_norm(x) [+ res]) on NHWC bf16 via the gfx950
        kernels in hip/bn_relu.hip. Replaces MIOpen's 5-pass forward /
        8-pass backward (incl. separate ReLU and residual-add kernels)
        with 3 forward passes and 5 (non-residual, mask recomputed
        from x) / 7 (residual, dym reuse) backward passes — see
        profiles/README.md for the motivation."""

        @staticmethod
        def forward(ctx, x, res, weight, bias, running_mean, running_var,
                    training, momentum, eps, relu):
            ext = _load()
            w32 = weight.float()
            b32 = bias.float()
            if training:
                y, mean, invstd = ext.bn_relu_fwd_train(
                    x, w32, b32, running_mean, running_var, momentum, eps,
                    res, relu)
                ctx.save_for_backward(x, y, w32, b32, mean, invstd)
                ctx.with_res = res is not None
                ctx.relu = relu
                ctx.wdtype = weight.dtype
                return y
            y = ext.bn_relu_fwd_eval(x, w32, b32, running_mean.float(),
                                     running_var.float(), eps, res, relu)
            ctx.save_for_backward(x, y, w32, b32, running_mean.float(),
                                  (running_var.float() + eps).rsqrt())
            ctx.with_res = res is not None
            ctx.relu = relu
            ctx.wdtype = weight.dtype
            return y

        @staticmethod
        def backward(ctx, dy):
            ext = _load()
            x, y, w32, b32, mean, invstd = ctx.saved_tensors
            out = ext.bn_relu_bwd(x, y, dy, w32, b32, mean, invstd,
                                  ctx.with_res, ctx.relu)
            dx, dscale, dbias = out[0], out[1], out[2]
            dres = out[3] if ctx.with_res else None
            return (dx, dres, dscale.to(ctx.wdtype), dbias.to(ctx.wdtype),
                    None, None, None, None, None, None)

    _BNReLUFn = BNReLUFn
    return BNReLUFn


_BNReLUFn = None


def bn_relu(x, bn, res=None, relu=True):
    """Fused BN[+ReLU][+residual add] using an nn.BatchNorm2d's
    parameters/buffers; relu=False serves activation-free BNs (the
    ResNet downsample path). Falls back to eager for shapes the kernel
    does not cover (non-bf16, C%8!=0, not channels-last)."""
    import torch

    supported = (x.dtype == torch.bfloat16 and x.size(1) % 8 == 0
                 and x.size(1) <= 2048 and x.is_cuda
                 and x.is_contiguous(memory_format=torch.channels_last))
    if supported and res is not None and res.dtype != torch.bfloat16:
        # downsample-path BN runs in fp32 under autocast; the residual
        # join is bf16 in bf16 training
        res = res.to(torch.bfloat16)
    if not supported:
        y = torch.nn.functional.batch_norm(
            x, bn.running_mean, bn.running_var, bn.weight, bn.bias,
            bn.training, bn.momentum, bn.eps)
        if res is not None:
            y = y + res
        return torch.relu(y) if relu else y
    if res is not None and not res.is_contiguous(
            memory_format=torch.channels_last):
        res = res.contiguous(memory_format=torch.channels_last)
    fn = _bn_relu_autograd()
    return fn.apply(x, res, bn.weight, bn.bias, bn.running_mean,
                    bn.running_var, bn.training, bn.momentum, bn.eps, relu)


def fuse_model(model):
    """Enable the fused BN+ReLU(+add) path on kubeshare_amd models
    (ResNet blocks check their `fused_ops` flag); verifies the extension
    is importable first (fail-loud on GPU boxes)."""
    _load()
    for m in model.modules():
        if hasattr(m, "fused_ops"):
            m.fused_ops = True
    return model
