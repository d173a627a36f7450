"""Fused multi-tensor Adam / SGD step (HIP kernel).

The fuse_optimizer pass (compiler/passes/fuse_optimizer.py) pattern-matches
the decomposed per-parameter Adam math in the sharded graph and replaces it
with ONE call to this op: a single grid-stride HIP kernel walking a chunk
table over every parameter shard (csrc/optim_kernels.hip) — the launch-bound
tail of the step collapses to one kernel.
"""
from __future__ import annotations

from typing import List

import torch

from . import load_extension

lib = torch.library.Library("easydist_amd", "FRAGMENT")
lib.define("fused_adam_step(Tensor[] params, Tensor[] grads, Tensor[] exp_avgs, "
           "Tensor[] exp_avg_sqs, Tensor[] steps, float lr, float beta1, "
           "float beta2, float weight_decay, float eps) "
           "-> (Tensor[], Tensor[], Tensor[], Tensor[])")


def _adam_aten(params, grads, exp_avgs, exp_avg_sqs, steps, lr, beta1, beta2,
               weight_decay, eps):
    new_p, new_ea, new_eas, new_steps = [], [], [], []
    for p, g, ea, eas, st in zip(params, grads, exp_avgs, exp_avg_sqs, steps):
        st = st + 1
        if weight_decay != 0:
            g = g + weight_decay * p
        ea = beta1 * ea + (1 - beta1) * g
        eas = beta2 * eas + (1 - beta2) * g * g
        bc1 = 1 - torch.pow(beta1, st)
        bc2 = 1 - torch.pow(beta2, st)
        denom = torch.sqrt(eas) / torch.sqrt(bc2) + eps
        p = p - lr * (ea / bc1) / denom
        new_p.append(p)
        new_ea.append(ea)
        new_eas.append(eas)
        new_steps.append(st)
    return new_p, new_ea, new_eas, new_steps


def _adam_cuda(params, grads, exp_avgs, exp_avg_sqs, steps, lr, beta1, beta2,
               weight_decay, eps):
    ext = load_extension()
    # capture-safe: the kernel stages its chunk table through persistent
    # pinned+device buffers sized by the eager warmup call, so hipGraph
    # capture records one async H2D + one kernel launch.  Grads may be
    # uniformly bf16 (the fuse pass folds the autocast fp32 casts).
    g_dtypes = {g.dtype for g in grads}
    if ext is not None and all(p.dtype == torch.float32 for p in params)             and (g_dtypes <= {torch.float32} or g_dtypes <= {torch.bfloat16}):
        return ext.fused_adam_step(list(params), list(grads), list(exp_avgs),
                                   list(exp_avg_sqs), list(steps), lr, beta1,
                                   beta2, weight_decay, eps)
    return _adam_aten(params, grads, exp_avgs, exp_avg_sqs, steps, lr, beta1,
                      beta2, weight_decay, eps)


lib.impl("fused_adam_step", _adam_aten, "CPU")
lib.impl("fused_adam_step", _adam_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::fused_adam_step")
def _adam_fake(params, grads, exp_avgs, exp_avg_sqs, steps, lr, beta1, beta2,
               weight_decay, eps):
    return ([torch.empty_like(p) for p in params],
            [torch.empty_like(t) for t in exp_avgs],
            [torch.empty_like(t) for t in exp_avg_sqs],
            [torch.empty_like(s) for s in steps])


lib.define("fused_sgd_step(Tensor[] params, Tensor[] grads, Tensor[] bufs, "
           "float lr, float momentum, float dampening, float weight_decay, "
           "bool nesterov) -> (Tensor[], Tensor[])")


def _sgd_aten(params, grads, bufs, lr, momentum, dampening, weight_decay,
              nesterov):
    new_p, new_b = [], []
    for i, (p, g) in enumerate(zip(params, grads)):
        if weight_decay != 0:
            g = g + weight_decay * p
        if momentum != 0:
            nb = momentum * bufs[i] + (1 - dampening) * g
            new_b.append(nb)
            g = g + momentum * nb if nesterov else nb
        new_p.append(p - lr * g)
    return new_p, new_b


def _sgd_cuda(params, grads, bufs, lr, momentum, dampening, weight_decay,
              nesterov):
    ext = load_extension()
    g_dtypes = {g.dtype for g in grads}
    if ext is not None and all(p.dtype == torch.float32 for p in params) \
            and (g_dtypes <= {torch.float32} or g_dtypes <= {torch.bfloat16}):
        return ext.fused_sgd_step(list(params), list(grads), list(bufs),
                                  lr, momentum, dampening, weight_decay,
                                  nesterov)
    return _sgd_aten(params, grads, bufs, lr, momentum, dampening,
                     weight_decay, nesterov)


lib.impl("fused_sgd_step", _sgd_aten, "CPU")
lib.impl("fused_sgd_step", _sgd_cuda, "CUDA")


@torch.library.register_fake("easydist_amd::fused_sgd_step")
def _sgd_fake(params, grads, bufs, lr, momentum, dampening, weight_decay,
              nesterov):
    return ([torch.empty_like(p) for p in params],
            [torch.empty_like(b) for b in bufs] if momentum != 0 else [])
