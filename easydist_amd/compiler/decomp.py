"""Decomposition table: fused/foreach optimizer ops -> plain aten + copy_.

Capability parity with reference ``easydist/torch/decomp_utils.py`` (lines
12-100): the whole-graph trace must contain only per-tensor functional math
so ShardCombine discovery and the sharding transform see ordinary ops. The
final ``copy_`` into each input buffer keeps train-state round-trip
semantics; the functionalize pass (passes/functionalize.py) later removes
them and rewires the graph outputs.

On MI355X the *runtime* counterpart of these decomposed optimizer ops is the
hand-written multi-tensor HIP Adam/SGD kernel (easydist_amd/ops/csrc/
optim_kernels.hip): the decomposition defines semantics + sharding; the
fuse_optimizer pass re-fuses the per-parameter math into one kernel launch
per step.
"""
import torch

aten = torch.ops.aten


def _single_adam(param, grad, exp_avg, exp_avg_sq, max_exp_avg_sq, step,
                 lr, beta1, beta2, weight_decay, eps, amsgrad, maximize):
    if maximize:
        grad = -grad
    if weight_decay != 0:
        grad = grad + weight_decay * param
    exp_avg = beta1 * exp_avg + (1 - beta1) * grad
    exp_avg_sq = beta2 * exp_avg_sq + (1 - beta2) * grad * grad
    bc1 = 1 - torch.pow(beta1, step)
    bc2 = 1 - torch.pow(beta2, step)
    if amsgrad:
        max_exp_avg_sq = torch.maximum(max_exp_avg_sq, exp_avg_sq)
        denom = torch.sqrt(max_exp_avg_sq) / torch.sqrt(bc2) + eps
    else:
        denom = torch.sqrt(exp_avg_sq) / torch.sqrt(bc2) + eps
    param = param - lr * (exp_avg / bc1) / denom
    return param, exp_avg, exp_avg_sq, max_exp_avg_sq


def fused_adam_decomp(params, grads, exp_avgs, exp_avg_sqs, max_exp_avg_sqs,
                      state_steps, *, lr, beta1, beta2, weight_decay, eps,
                      amsgrad, maximize, grad_scale=None, found_inf=None):
    for i in range(len(params)):
        mx = max_exp_avg_sqs[i] if amsgrad else None
        g = grads[i]
        if grad_scale is not None:
            g = g / grad_scale
        p, ea, eas, mx2 = _single_adam(params[i], g, exp_avgs[i],
                                       exp_avg_sqs[i], mx, state_steps[i], lr,
                                       beta1, beta2, weight_decay, eps,
                                       amsgrad, maximize)
        params[i].copy_(p)
        exp_avgs[i].copy_(ea)
        exp_avg_sqs[i].copy_(eas)
        if amsgrad:
            max_exp_avg_sqs[i].copy_(mx2)


def fused_adamw_decomp(params, grads, exp_avgs, exp_avg_sqs, max_exp_avg_sqs,
                       state_steps, *, lr, beta1, beta2, weight_decay, eps,
                       amsgrad, maximize, grad_scale=None, found_inf=None):
    for i in range(len(params)):
        g = grads[i]
        if grad_scale is not None:
            g = g / grad_scale
        if maximize:
            g = -g
        p = params[i] * (1 - lr * weight_decay)
        ea = beta1 * exp_avgs[i] + (1 - beta1) * g
        eas = beta2 * exp_avg_sqs[i] + (1 - beta2) * g * g
        bc1 = 1 - torch.pow(beta1, state_steps[i])
        bc2 = 1 - torch.pow(beta2, state_steps[i])
        if amsgrad:
            mx = torch.maximum(max_exp_avg_sqs[i], eas)
            denom = torch.sqrt(mx) / torch.sqrt(bc2) + eps
            max_exp_avg_sqs[i].copy_(mx)
        else:
            denom = torch.sqrt(eas) / torch.sqrt(bc2) + eps
        p = p - lr * (ea / bc1) / denom
        params[i].copy_(p)
        exp_avgs[i].copy_(ea)
        exp_avg_sqs[i].copy_(eas)


def fused_sgd_decomp(params, grads, momentum_buffer_list, *, weight_decay,
                     momentum, lr, dampening, nesterov, maximize, is_first_step=False,
                     grad_scale=None, found_inf=None):
    for i in range(len(params)):
        g = grads[i]
        if grad_scale is not None:
            g = g / grad_scale
        if maximize:
            g = -g
        if weight_decay != 0:
            g = g + weight_decay * params[i]
        if momentum != 0:
            buf = momentum_buffer_list[i]
            if is_first_step:
                nb = g
            else:
                nb = momentum * buf + (1 - dampening) * g
            momentum_buffer_list[i].copy_(nb)
            g = g + momentum * nb if nesterov else nb
        params[i].copy_(params[i] - lr * g)


# ------------------------------------------------ foreach functionalization --
def _foreach_binary_inplace(op):
    def decomp(self, other, alpha=None, **kw):
        if isinstance(other, (list, tuple)):
            for t, o in zip(self, other):
                t.copy_(op(t, o, alpha) if alpha is not None else op(t, o, None))
        else:
            for t in self:
                t.copy_(op(t, other, alpha) if alpha is not None else op(t, other, None))
    return decomp


def _add(a, b, alpha):
    return a + (b * alpha if alpha is not None else b)


def _mul(a, b, _):
    return a * b


def _div(a, b, _):
    return a / b


def _sub(a, b, alpha):
    return a - (b * alpha if alpha is not None else b)


EASYDIST_DECOMP_TABLE = {
    aten._fused_adam_.default: fused_adam_decomp,
    aten._fused_adamw_.default: fused_adamw_decomp,
    aten._fused_sgd_.default: fused_sgd_decomp,
    aten._foreach_add_.Scalar: _foreach_binary_inplace(_add),
    aten._foreach_add_.List: _foreach_binary_inplace(_add),
    aten._foreach_mul_.Scalar: _foreach_binary_inplace(_mul),
    aten._foreach_mul_.List: _foreach_binary_inplace(_mul),
    aten._foreach_div_.Scalar: _foreach_binary_inplace(_div),
    aten._foreach_div_.List: _foreach_binary_inplace(_div),
    aten._foreach_sub_.Scalar: _foreach_binary_inplace(_sub),
    aten._foreach_sub_.List: _foreach_binary_inplace(_sub),
}
