"""LayerNorm / RMSNorm as custom ops backed by hand HIP kernels.

The lowering pass (compiler/passes/lower_hip.py) rewrites
aten.native_layer_norm(_backward) nodes in the sharded graph to these ops,
so an unmodified user model still runs the gfx950 kernels
(csrc/norm_kernels.hip: one-pass Welford fwd, two-pass bwd, bf16x8
vectorized loads — memory-bound ops tuned for the 8 TB/s HBM3E roofline).
"""
from __future__ import annotations

import torch

from . import load_extension

lib = torch.library.Library("easydist_amd", "FRAGMENT")
lib.define("layer_norm_fwd(Tensor x, Tensor? w, Tensor? b, float eps) "
           "-> (Tensor, Tensor, Tensor)")
lib.define("layer_norm_bwd(Tensor grad, Tensor x, Tensor mean, Tensor rstd, "
           "Tensor? w, bool[3] mask) -> (Tensor, Tensor, Tensor)")
lib.define("ln_bwd_ref(Tensor grad, Tensor x, SymInt[] norm_shape, "
           "Tensor mean, Tensor rstd, Tensor? w, Tensor? b, bool[] mask) "
           "-> (Tensor, Tensor, Tensor)")
lib.define("rms_norm_fwd(Tensor x, Tensor? w, float eps) -> (Tensor, Tensor)")
lib.define("rms_norm_bwd(Tensor grad, Tensor x, Tensor rstd, Tensor? w) "
           "-> (Tensor, Tensor)")


def _ln_fwd_aten(x, w, b, eps):
    d = x.shape[-1]
    out, mean, rstd = torch.ops.aten.native_layer_norm(x, [d], w, b, eps)
    return out, mean, rstd


def _ln_fwd_cuda(x, w, b, eps):
    ext = load_extension()
    if ext is not None and x.dtype in (torch.bfloat16, torch.float32) \
            and x.shape[-1] % 8 == 0:
        # the kernel reads w/b as x's dtype
        w = w.to(x.dtype).contiguous() if w is not None else None
        b = b.to(x.dtype).contiguous() if b is not None else None
        return ext.layer_norm_fwd(x.contiguous(), w, b, eps)
    return _ln_fwd_aten(x, w, b, eps)


def _ln_bwd_ref(grad, x, norm_shape, mean, rstd, w, b, mask):
    """Dtype-tolerant native_layer_norm_backward: the sharded graph can
    hand the un-lowered aten node mixed fp32/bf16 args after cast
    folding around its (lowered) forward — aten rejects that. Compute
    fully in fp32 and let the lowering pass cast outputs back to the
    dtypes the graph's meta expects."""
    f = lambda t: None if t is None else t.float().contiguous()
    n = len(norm_shape)
    # stats have one entry per non-normalized row: [*batch_dims, 1...]
    ms = [int(d) for d in grad.shape[:grad.dim() - n]] + [1] * n
    outs = torch.ops.aten.native_layer_norm_backward.default(
        f(grad), f(x), list(norm_shape),
        f(mean).reshape(ms), f(rstd).reshape(ms),
        f(w), f(b), list(mask))
    z = lambda k, like: (outs[k] if outs[k] is not None
                         else torch.zeros_like(like, dtype=torch.float32))
    dw_like = w if w is not None else grad.new_empty(norm_shape)
    return (outs[0] if outs[0] is not None else torch.zeros_like(grad,
                                                                 dtype=torch.float32),
            z(1, dw_like), z(2, dw_like))


lib.impl("ln_bwd_ref", _ln_bwd_ref, "CompositeExplicitAutograd")


@torch.library.register_fake("easydist_amd::ln_bwd_ref")
def _ln_bwd_ref_fake(grad, x, norm_shape, mean, rstd, w, b, mask):
    shp = list(norm_shape)
    return (torch.empty_like(grad, dtype=torch.float32),
            torch.empty(shp, dtype=torch.float32, device=grad.device),
            torch.empty(shp, dtype=torch.float32, device=grad.device))


def _ln_bwd_aten(grad, x, mean, rstd, w, mask):
    """Emulates the HIP kernel contract on aten: fp32 math, dx in x's
    dtype, dw/db fp32 (the CPU aten kernel also rejects mixed
    bf16-input/fp32-stat calls that CUDA accepts)."""
    d = x.shape[-1]
    bias = (torch.zeros(d, dtype=torch.float32, device=x.device)
            if len(mask) > 2 and mask[2] else None)
    wf = w.float() if w is not None else None
    dx, dw, db = torch.ops.aten.native_layer_norm_backward(
        grad.float(), x.float(), [d], mean.float(), rstd.float(), wf, bias,
        list(mask))
    return (dx.to(x.dtype) if dx is not None else None, dw, db)


def _ln_bwd_cuda(grad, x, mean, rstd, w, mask):
    ext = load_extension()
    if ext is not None and x.dtype == torch.bfloat16 \
            and x.shape[-1] % 8 == 0:
        w2 = w.to(x.dtype).contiguous() if w is not None else None
        return ext.layer_norm_bwd(grad.to(x.dtype).contiguous(),
                                  x.contiguous(), mean.contiguous(),
                                  rstd.contiguous(), w2, list(mask))
    return _ln_bwd_aten(grad, x, mean, rstd, w, mask)


lib.impl("layer_norm_fwd", _ln_fwd_aten, "CPU")
lib.impl("layer_norm_fwd", _ln_fwd_cuda, "CUDA")
lib.impl("layer_norm_bwd", _ln_bwd_aten, "CPU")
lib.impl("layer_norm_bwd", _ln_bwd_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::layer_norm_fwd")
def _ln_fwd_fake(x, w, b, eps):
    stat_shape = x.shape[:-1] + (1,)
    stat = x.new_empty(stat_shape, dtype=torch.float32)
    return torch.empty_like(x), stat, stat.clone()


@torch.library.register_fake("easydist_amd::layer_norm_bwd")
def _ln_bwd_fake(grad, x, mean, rstd, w, mask):
    d = x.shape[-1]
    dw = x.new_empty((d,), dtype=w.dtype if w is not None else x.dtype)
    return torch.empty_like(x), dw, dw.clone()


def _rms_fwd_aten(x, w, eps):
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    out = xf * rstd
    if w is not None:
        out = out * w.float()
    return out.to(x.dtype), rstd


def _rms_fwd_cuda(x, w, eps):
    ext = load_extension()
    if ext is not None and x.dtype in (torch.bfloat16, torch.float32) \
            and x.shape[-1] % 8 == 0:
        return ext.rms_norm_fwd(x.contiguous(),
                                w.contiguous() if w is not None else None, eps)
    return _rms_fwd_aten(x, w, eps)


def _rms_bwd_aten(grad, x, rstd, w):
    xf, gf = x.float(), grad.float()
    if w is not None:
        gw = (gf * (xf * rstd)).sum(
            tuple(range(x.dim() - 1)))
        gf = gf * w.float()
    else:
        gw = torch.zeros(x.shape[-1], device=x.device)
    d = x.shape[-1]
    xhat = xf * rstd
    dx = rstd * (gf - xhat * (gf * xhat).mean(-1, keepdim=True))
    return dx.to(x.dtype), gw.to(w.dtype if w is not None else x.dtype)


def _rms_bwd_cuda(grad, x, rstd, w):
    ext = load_extension()
    if ext is not None and x.dtype in (torch.bfloat16, torch.float32) \
            and x.shape[-1] % 8 == 0:
        return ext.rms_norm_bwd(grad.contiguous(), x.contiguous(), rstd,
                                w.contiguous() if w is not None else None)
    return _rms_bwd_aten(grad, x, rstd, w)


lib.impl("rms_norm_fwd", _rms_fwd_aten, "CPU")
lib.impl("rms_norm_fwd", _rms_fwd_cuda, "CUDA")
lib.impl("rms_norm_bwd", _rms_bwd_aten, "CPU")
lib.impl("rms_norm_bwd", _rms_bwd_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::rms_norm_fwd")
def _rms_fwd_fake(x, w, eps):
    stat = x.new_empty(x.shape[:-1] + (1,), dtype=torch.float32)
    return torch.empty_like(x), stat


@torch.library.register_fake("easydist_amd::rms_norm_bwd")
def _rms_bwd_fake(grad, x, rstd, w):
    d = x.shape[-1]
    return torch.empty_like(x), x.new_empty((d,))


class RMSNorm(torch.nn.Module):
    def __init__(self, dim, eps=1e-6):
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        out, _ = torch.ops.easydist_amd.rms_norm_fwd(x, self.weight, self.eps)
        return out
