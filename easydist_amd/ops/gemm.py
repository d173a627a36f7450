"""Hand-written MFMA GEMM as custom ops, with per-shape profiled dispatch.

The lowering pass (compiler/passes/lower_hip.py:lower_gemm) rewrites the
sharded graph's linear-layer matmuls to these ops:

* ``easydist_amd::gemm_nt(a, bt, bias)`` — C = a @ bt.T (+bias), the
  NT shape every traced ``nn.Linear`` forward produces (both operands
  K-contiguous), and — with a cheap weight-side transpose inserted by the
  pass — the dX backward too.  CUDA: the 256x256 8-phase MFMA kernel
  (csrc/gemm_kernels.hip), 128x128 for edge shapes.
* ``easydist_amd::gemm_tn(a, b)`` — C = a.T @ b with BOTH operands
  reduce-dim-strided (the dW backward; transposing the activations
  globally would cost about the GEMM itself).  CUDA: split-R tr-read
  MFMA kernel.

Dispatch policy (EASYDIST_GEMM_POLICY): ``hand`` always uses the HIP
kernel when the shape is supported; ``aten`` never does (hipBLASLt);
``auto`` (default) microbenchmarks both on first sight of a shape and
caches the winner, preferring the hand kernel within a 3% tie margin
(reference VERDICT item 1: per-shape profiled fallback).  The autotuner
refuses to run while a hipGraph capture is active (it synchronizes) —
shapes must be warmed up eagerly first, which the compile warmup does.
"""
from __future__ import annotations

import logging
import os
from typing import Optional

import torch

from . import load_extension
from .. import config as mdconfig

logger = logging.getLogger(__name__)

lib = torch.library.Library("easydist_amd", "FRAGMENT")
lib.define("gemm_nt(Tensor a, Tensor bt, Tensor? bias) -> Tensor")
lib.define("gemm_tn(Tensor a, Tensor b) -> Tensor")


def _policy() -> str:
    return os.environ.get("EASYDIST_GEMM_POLICY", "auto")


# ---------------------------------------------------------------- CPU -------
def _nt_cpu(a, bt, bias):
    # addmm (not mm+add): hipBLASLt fuses the bias into its epilogue —
    # the separate broadcast add costs an output-sized HBM round-trip
    if bias is not None:
        return torch.addmm(bias, a, bt.t())
    return a @ bt.t()


def _tn_cpu(a, b):
    return a.t() @ b


# ---------------------------------------------------------------- CUDA ------
def _nt_supported(a, bt):
    return (a.dtype == torch.bfloat16 and bt.dtype == torch.bfloat16
            and a.dim() == 2 and bt.dim() == 2
            and a.shape[1] == bt.shape[1]
            and a.shape[1] % 32 == 0 and bt.shape[0] % 8 == 0
            and a.shape[0] >= 16)


def _tn_supported(a, b):
    return (a.dtype == torch.bfloat16 and b.dtype == torch.bfloat16
            and a.dim() == 2 and b.dim() == 2 and a.shape[0] == b.shape[0]
            and a.shape[0] % 32 == 0
            and a.shape[1] % 128 == 0 and b.shape[1] % 128 == 0)


_tune_cache = {}


def _time_fn(fn, iters=5):
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    fn(); fn()  # warm
    torch.cuda.synchronize()
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    end.synchronize()
    return start.elapsed_time(end) / iters


def _choose(key, hand_fn, aten_fn):
    """Per-shape profiled dispatch. Returns True to use the hand kernel."""
    pol = _policy()
    if pol == "hand":
        return True
    if pol == "aten":
        return False
    if key in _tune_cache:
        return _tune_cache[key]
    if torch.cuda.is_current_stream_capturing():
        # cannot time inside capture; the warmup should have seeded the
        # cache — default to the hand kernel rather than silently
        # switching the captured graph's kernels
        return True
    try:
        t_hand = _time_fn(hand_fn)
        t_aten = _time_fn(aten_fn)
        use = t_hand <= t_aten * 1.03
        logger.info("gemm autotune %s: hand %.3fms aten %.3fms -> %s",
                    key, t_hand, t_aten, "hand" if use else "aten")
    except Exception as e:  # pragma: no cover
        logger.warning("gemm autotune failed for %s: %s", key, e)
        use = False
    _tune_cache[key] = use
    return use


def _nt_cuda(a, bt, bias):
    ext = load_extension()
    if ext is None or not _nt_supported(a, bt) \
            or not mdconfig.use_hip_kernels:
        if ext is None and mdconfig.use_hip_kernels:
            from . import require_hip_ops
            require_hip_ops()
        return _nt_cpu(a, bt, bias)
    a = a.contiguous()
    bt = bt.contiguous()
    b_c = bias.contiguous() if bias is not None else None
    key = ("nt", a.shape[0], a.shape[1], bt.shape[0], bias is not None)
    if _choose(key, lambda: ext.gemm_nt(a, bt, b_c),
               lambda: _nt_cpu(a, bt, bias)):
        return ext.gemm_nt(a, bt, b_c)
    return _nt_cpu(a, bt, bias)


def _tn_cuda(a, b):
    ext = load_extension()
    if ext is None or not _tn_supported(a, b) \
            or not mdconfig.use_hip_kernels:
        if ext is None and mdconfig.use_hip_kernels:
            from . import require_hip_ops
            require_hip_ops()
        return _tn_cpu(a, b)
    a = a.contiguous()
    b = b.contiguous()
    key = ("tn", a.shape[0], a.shape[1], b.shape[1])
    if _choose(key, lambda: ext.gemm_tn(a, b), lambda: _tn_cpu(a, b)):
        return ext.gemm_tn(a, b)
    return _tn_cpu(a, b)


lib.impl("gemm_nt", _nt_cpu, "CPU")
lib.impl("gemm_nt", _nt_cuda, "CUDA")
lib.impl("gemm_tn", _tn_cpu, "CPU")
lib.impl("gemm_tn", _tn_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::gemm_nt")
def _nt_fake(a, bt, bias):
    return a.new_empty((a.shape[0], bt.shape[0]))


@torch.library.register_fake("easydist_amd::gemm_tn")
def _tn_fake(a, b):
    return a.new_empty((a.shape[1], b.shape[1]))


# ---------------------------------------------------------- fused variants --
lib.define("gemm_nt_act(Tensor a, Tensor bt, Tensor? bias, int act, "
           "Tensor? aux) -> Tensor")
lib.define("gemm_tn_asum(Tensor a, Tensor b) -> (Tensor, Tensor)")

_GELU_TANH, _GELU_TANH_BWD, _GELU_ERF, _GELU_ERF_BWD = 1, 2, 3, 4


def _apply_act(out, act, aux):
    if act == _GELU_TANH:
        return torch.nn.functional.gelu(out, approximate="tanh")
    if act == _GELU_ERF:
        return torch.nn.functional.gelu(out)
    if act == _GELU_TANH_BWD:
        return torch.ops.aten.gelu_backward(out, aux, approximate="tanh")
    if act == _GELU_ERF_BWD:
        return torch.ops.aten.gelu_backward(out, aux, approximate="none")
    return out


def _nt_act_cpu(a, bt, bias, act, aux):
    return _apply_act(_nt_cpu(a, bt, bias), act, aux)


def _act_fast(ext, out, act, aux):
    """Apply the activation with the bandwidth-rate HIP kernel (aten's
    gelu runs at ~4.6 TB/s on MLP activations — erf libcall latency
    chains); erf modes use the same tanh approximation the fused
    epilogue uses (<3.2e-3 difference, under bf16 resolution)."""
    if ext is not None and out.dtype == torch.bfloat16 \
            and out.numel() % 8 == 0:
        if act in (_GELU_TANH, _GELU_ERF):
            return ext.gelu_fast(out)
        if act in (_GELU_TANH_BWD, _GELU_ERF_BWD):
            return ext.gelu_bwd_fast(out, aux.contiguous())
    return _apply_act(out, act, aux)


def _nt_act_aten(ext, a, bt, bias, act, aux):
    return _act_fast(ext, _nt_cpu(a, bt, bias), act, aux)


def _nt_act_cuda(a, bt, bias, act, aux):
    ext = load_extension()
    if ext is None or not _nt_supported(a, bt) \
            or not mdconfig.use_hip_kernels:
        return _nt_act_cpu(a, bt, bias, act, aux)
    a = a.contiguous()
    bt = bt.contiguous()
    b_c = bias.contiguous() if bias is not None else None
    x_c = aux.contiguous() if aux is not None else None
    # the comparison must price the FUSION: choosing aten re-adds the
    # separate activation kernel the epilogue absorbs
    key = ("nt", a.shape[0], a.shape[1], bt.shape[0], bias is not None, act)
    if _choose(key, lambda: ext.gemm_nt_act(a, bt, b_c, act, x_c),
               lambda: _nt_act_aten(ext, a, bt, bias, act, aux)):
        return ext.gemm_nt_act(a, bt, b_c, act, x_c)
    return _nt_act_aten(ext, a, bt, bias, act, aux)


def _tn_asum_cpu(a, b):
    # dtype=... keeps the reduction fused — a.float() would materialize a
    # full fp32 copy of the activation-sized gradient
    return a.t() @ b, a.sum(0, dtype=torch.float32)


def _tn_asum_cuda(a, b):
    ext = load_extension()
    if ext is None or not _tn_supported(a, b) \
            or not mdconfig.use_hip_kernels:
        return _tn_asum_cpu(a, b)
    a = a.contiguous()
    b = b.contiguous()
    key = ("tn_asum", a.shape[0], a.shape[1], b.shape[1])
    if _choose(key, lambda: ext.gemm_tn_asum(a, b),
               lambda: _tn_asum_cpu(a, b)):
        return ext.gemm_tn_asum(a, b)
    return _tn_asum_cpu(a, b)


lib.impl("gemm_nt_act", _nt_act_cpu, "CPU")
lib.impl("gemm_nt_act", _nt_act_cuda, "CUDA")
lib.impl("gemm_tn_asum", _tn_asum_cpu, "CPU")
lib.impl("gemm_tn_asum", _tn_asum_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::gemm_nt_act")
def _nt_act_fake(a, bt, bias, act, aux):
    return a.new_empty((a.shape[0], bt.shape[0]))


@torch.library.register_fake("easydist_amd::gemm_tn_asum")
def _tn_asum_fake(a, b):
    return (a.new_empty((a.shape[1], b.shape[1])),
            a.new_empty((a.shape[1],), dtype=torch.float32))


# fused forward linear+gelu: one kernel writes BOTH gelu(out) and the
# pre-activation (kept for gelu_backward), saving the standalone
# activation kernel's HBM round-trip (lower_hip.py:lower_gelu_fwd_fuse)
lib.define("gemm_nt_gelu(Tensor a, Tensor bt, Tensor? bias, bool tanh_approx)"
           " -> (Tensor, Tensor)")


def _nt_gelu_cpu(a, bt, bias, tanh_approx):
    pre = _nt_cpu(a, bt, bias)
    out = torch.nn.functional.gelu(
        pre, approximate="tanh" if tanh_approx else "none")
    return out, pre


def _nt_gelu_aten(ext, a, bt, bias, tanh_approx):
    pre = _nt_cpu(a, bt, bias)
    act = _GELU_TANH if tanh_approx else _GELU_ERF
    return _act_fast(ext, pre, act, None), pre


def _nt_gelu_cuda(a, bt, bias, tanh_approx):
    ext = load_extension()
    if ext is None or not _nt_supported(a, bt) \
            or not mdconfig.use_hip_kernels:
        if ext is None and mdconfig.use_hip_kernels:
            from . import require_hip_ops
            require_hip_ops()
        return _nt_gelu_cpu(a, bt, bias, tanh_approx)
    a = a.contiguous()
    bt = bt.contiguous()
    b_c = bias.contiguous() if bias is not None else None
    key = ("nt_gelu", a.shape[0], a.shape[1], bt.shape[0], bias is not None,
           tanh_approx)
    if _choose(key, lambda: ext.gemm_nt_gelu(a, bt, b_c, tanh_approx),
               lambda: _nt_gelu_aten(ext, a, bt, bias, tanh_approx)):
        return ext.gemm_nt_gelu(a, bt, b_c, tanh_approx)
    return _nt_gelu_aten(ext, a, bt, bias, tanh_approx)


lib.impl("gemm_nt_gelu", _nt_gelu_cpu, "CPU")
lib.impl("gemm_nt_gelu", _nt_gelu_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::gemm_nt_gelu")
def _nt_gelu_fake(a, bt, bias, tanh_approx):
    return (a.new_empty((a.shape[0], bt.shape[0])),
            a.new_empty((a.shape[0], bt.shape[0])))


# NN-layout GEMM: C = a @ b with b [K, N] row-major (the dX backward:
# dX = dY @ W). The HAND kernel wants a K-contiguous B operand, so ITS
# route pays one weight-sized transpose copy; the hipBLASLt route takes
# the strided layout natively and skips it. Keeping the transpose INSIDE
# the op lets the per-shape profiler price each route honestly — the
# old lowering cloned t(W) in the graph unconditionally.
lib.define("gemm_nn(Tensor a, Tensor b, Tensor? bias) -> Tensor")


def _nn_cpu(a, b, bias):
    if bias is not None:
        return torch.addmm(bias, a, b)
    return a @ b


def _nn_cuda(a, b, bias):
    ext = load_extension()
    if ext is None or not mdconfig.use_hip_kernels or a.dim() != 2 \
            or b.dim() != 2 or a.dtype != torch.bfloat16 \
            or b.dtype != torch.bfloat16 \
            or not _nt_supported(a, b.t()):   # shape check only (view)
        return _nn_cpu(a, b, bias)
    a = a.contiguous()

    def hand():
        # the hand kernel's route INCLUDES its weight transpose copy, so
        # the profiler prices it honestly against the strided aten mm
        bt = b.t().contiguous()
        b_c = bias.contiguous() if bias is not None else None
        return ext.gemm_nt(a, bt, b_c)

    key = ("nn", a.shape[0], a.shape[1], b.shape[1], bias is not None)
    if _choose(key, hand, lambda: _nn_cpu(a, b, bias)):
        return hand()
    return _nn_cpu(a, b, bias)


lib.impl("gemm_nn", _nn_cpu, "CPU")
lib.impl("gemm_nn", _nn_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::gemm_nn")
def _nn_fake(a, b, bias):
    return a.new_empty((a.shape[0], b.shape[1]))


# gemm_nn with a fused activation epilogue (the dX + gelu_backward
# fusion when the incoming grad is an NN-layout GEMM): hand route =
# weight transpose + gemm_nt_act; aten route = strided mm + the
# bandwidth-rate gelu_bwd kernel.
lib.define("gemm_nn_act(Tensor a, Tensor b, Tensor? bias, int act, "
           "Tensor? aux) -> Tensor")


def _nn_act_cpu(a, b, bias, act, aux):
    return _apply_act(_nn_cpu(a, b, bias), act, aux)


def _nn_act_cuda(a, b, bias, act, aux):
    ext = load_extension()
    if ext is None or not mdconfig.use_hip_kernels or a.dim() != 2 \
            or b.dim() != 2 or a.dtype != torch.bfloat16 \
            or b.dtype != torch.bfloat16 \
            or not _nt_supported(a, b.t()):
        return _nn_act_cpu(a, b, bias, act, aux)
    a = a.contiguous()
    x_c = aux.contiguous() if aux is not None else None

    def hand():
        bt = b.t().contiguous()
        b_c = bias.contiguous() if bias is not None else None
        return ext.gemm_nt_act(a, bt, b_c, act, x_c)

    def aten():
        return _act_fast(ext, _nn_cpu(a, b, bias), act, aux)

    key = ("nn", a.shape[0], a.shape[1], b.shape[1], bias is not None, act)
    if _choose(key, hand, aten):
        return hand()
    return aten()


lib.impl("gemm_nn_act", _nn_act_cpu, "CPU")
lib.impl("gemm_nn_act", _nn_act_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::gemm_nn_act")
def _nn_act_fake(a, b, bias, act, aux):
    return a.new_empty((a.shape[0], b.shape[1]))
