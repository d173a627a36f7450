"""Re-fuse the decomposed Adam math into one multi-tensor kernel call.

The whole-step trace decomposes ``aten._fused_adam_`` into per-parameter
elementwise chains (decomp.py) so ShardCombine discovery and the solver
see ordinary ops; after the sharding transform those chains are a
launch-bound tail of hundreds of tiny elementwise kernels (rocprof:
fp32 mul/add ~12% of step time on GPT-2). This pass pattern-matches each
parameter's chain and replaces ALL of them with ONE
``easydist_amd::fused_adam_step`` call — a single grid-stride HIP kernel
walking a chunk table over every parameter shard
(ops/csrc/optim_kernels.hip).

Safe-by-construction: if any parameter's chain does not match the
expected decomposition exactly, the pass fuses the matching subset only
(or nothing) and leaves the graph semantics untouched.
"""
from __future__ import annotations

import logging
import operator
from typing import Dict, List, Optional, Tuple

import torch
import torch.fx as fx

from ...ops import optim as _optim  # noqa: F401  (registers the custom op)

logger = logging.getLogger(__name__)

aten = torch.ops.aten


def _tensor_scalar(n) -> Tuple[Optional[fx.Node], Optional[float]]:
    """mul.Tensor(tensor, scalar) in either arg order."""
    if not (isinstance(n, fx.Node) and n.op == "call_function"
            and n.target is aten.mul.Tensor):
        return None, None
    a, b = n.args
    if isinstance(a, fx.Node) and isinstance(b, (int, float)):
        return a, float(b)
    if isinstance(b, fx.Node) and isinstance(a, (int, float)):
        return b, float(a)
    return None, None


def _strip_l2(g, p_ph, wd):
    """Coupled (Adam) weight decay traces as g' = add(raw, mul(p, wd))
    (decomp.py:_single_adam). Return the RAW grad node — the kernel
    re-applies ``grad += wd*p`` itself — or None if the shape differs."""
    if not (isinstance(g, fx.Node) and g.target is aten.add.Tensor):
        return None
    for a, b in (g.args, g.args[::-1]):
        t, s = _tensor_scalar(b)
        if t is p_ph and s is not None and abs(s - wd) < 1e-12 \
                and isinstance(a, fx.Node):
            return a
    return None


def _is_scaled_square(n, g) -> bool:
    """Match s·g·g in any association: mul(mul(g, s), g), mul(g, mul(g, s))
    or mul(mul(g, g), s)."""
    if not (isinstance(n, fx.Node) and n.op == "call_function"
            and n.target is aten.mul.Tensor):
        return False
    a, b = n.args
    for x, y in ((a, b), (b, a)):
        if y is g:
            t, s = _tensor_scalar(x)
            if t is g and s is not None:
                return True
        if isinstance(y, (int, float)) and isinstance(x, fx.Node) \
                and x.op == "call_function" and x.target is aten.mul.Tensor \
                and x.args[0] is g and x.args[1] is g:
            return True
    return False


def _match_param_chain(p_ph, ea_ph, eas_ph, step_ph, p_new, ea_new,
                       eas_new, step_new, wd=0.0, decay_scale=None):
    """Verify the decomposed Adam/AdamW shape; return the grad node.

    ``wd``: coupled L2 decay (Adam) — the grad feeding the moments is
    add(raw_g, wd*p); we unwrap it. ``decay_scale``: decoupled decay
    (AdamW) — p_new subtracts from mul(p_ph, 1-lr*wd) instead of p_ph.
    """
    # step_new = add(step_ph, 1)
    if not (isinstance(step_new, fx.Node)
            and step_new.target in (aten.add.Tensor, aten.add.Scalar)
            and step_new.args[0] is step_ph):
        return None
    # ea_new = add(mul(ea_ph, b1), mul(g, 1-b1))
    if not (isinstance(ea_new, fx.Node)
            and ea_new.target is aten.add.Tensor):
        return None
    t0, s0 = _tensor_scalar(ea_new.args[0])
    t1, s1 = _tensor_scalar(ea_new.args[1])
    if t0 is None or t1 is None:
        return None
    if t0 is ea_ph:
        g = t1
    elif t1 is ea_ph:
        g = t0
    else:
        return None
    # the eas chain squares the (possibly L2-decayed) grad node; keep it
    # before unwrapping the decay for the return value
    g_dec = g
    if wd != 0.0:
        g = _strip_l2(g, p_ph, wd)
        if g is None:
            return None
    # eas_new = add(mul(eas_ph, b2), mul(mul(g, 1-b2), g)) — verify BOTH
    # operands, same as the ea chain: one term scales eas_ph, the other
    # is the scaled square of the same grad node.
    if not (isinstance(eas_new, fx.Node)
            and eas_new.target is aten.add.Tensor):
        return None
    q0, q1 = eas_new.args
    qt0, _ = _tensor_scalar(q0)
    qt1, _ = _tensor_scalar(q1)
    if qt0 is eas_ph:
        g2 = q1
    elif qt1 is eas_ph:
        g2 = q0
    else:
        return None
    if not _is_scaled_square(g2, g_dec):
        return None
    # p_new = sub(base, ...); base = p_ph (Adam) or mul(p_ph, 1-lr*wd)
    # (AdamW decomp always emits the mul, even at wd=0 -> scalar 1.0)
    if not (isinstance(p_new, fx.Node)
            and p_new.target is aten.sub.Tensor):
        return None
    base = p_new.args[0]
    if decay_scale is not None:
        t, s = _tensor_scalar(base)
        if not (t is p_ph and s is not None
                and abs(s - decay_scale) < 1e-12):
            return None
    elif base is not p_ph:
        return None
    return g


def fuse_optimizer(gm: fx.GraphModule, flat_outs: List, placeholders: List,
                   param_positions: Dict[int, Dict[str, int]],
                   opt, pl_env=None) -> int:
    """Fuse matching per-param Adam chains. Returns #fused params.

    param_positions: param flat-input position -> {'step': out_pos,
    'exp_avg': out_pos, 'exp_avg_sq': out_pos, 'param': out_pos}
    (positions into flat_outs; the same indexes are placeholder input
    positions for the state tensors).
    """
    if opt is None:
        return 0
    if type(opt).__name__ == "SGD":
        return _fuse_sgd(gm, flat_outs, placeholders, param_positions, opt,
                         pl_env)
    if type(opt).__name__ not in ("Adam", "AdamW"):
        return 0
    decoupled = type(opt).__name__ == "AdamW"
    groups = opt.param_groups
    if len(groups) != 1 or groups[0].get("amsgrad") \
            or groups[0].get("maximize"):
        return 0
    lr = float(groups[0]["lr"])
    beta1, beta2 = map(float, groups[0]["betas"])
    eps = float(groups[0]["eps"])
    wd = float(groups[0].get("weight_decay", 0.0))
    # decoupled (AdamW): p is pre-scaled by (1-lr*wd) before the Adam
    # update; we emit ONE _foreach_mul for the whole bank and hand the
    # kernel wd=0. Coupled (Adam): the kernel applies grad += wd*p itself.
    decay_scale = (1.0 - lr * wd) if decoupled else None
    kernel_wd = 0.0 if decoupled else wd
    chain_wd = 0.0 if decoupled else wd

    def _all_replicate(*nodes):
        """The fused call wires these nodes DIRECTLY. Safe when every
        placement is replicate, OR when ALL of them carry the SAME
        placement vector: the Adam update is elementwise, so a uniform
        sharding of param/grad/moments is exactly the local update on
        each shard (the ZeRO-like assignments the solver picks at
        world>1). Mixed placements would need the reshards the direct
        wiring bypasses — rejected."""
        if not pl_env:
            return True
        seen = []
        for n in nodes:
            pls = pl_env.get(n.name)
            if not pls:
                continue
            seen.append(tuple(repr(p) for p in pls[0]))
        if not seen:
            return True
        if all(all(p == "R" for p in v) for v in seen):
            return True
        return len(set(seen)) == 1 and len(seen) == len(nodes)

    matched = []
    for p_pos, outs in param_positions.items():
        p_ph = placeholders[p_pos]
        ea_ph = placeholders[outs["exp_avg_in"]]
        eas_ph = placeholders[outs["exp_avg_sq_in"]]
        step_ph = placeholders[outs["step_in"]]
        p_new = flat_outs[outs["param"]]
        ea_new = flat_outs[outs["exp_avg"]]
        eas_new = flat_outs[outs["exp_avg_sq"]]
        step_new = flat_outs[outs["step"]]
        g = _match_param_chain(p_ph, ea_ph, eas_ph, step_ph, p_new, ea_new,
                               eas_new, step_new, wd=chain_wd,
                               decay_scale=decay_scale)
        if g is None:
            continue
        if not _all_replicate(p_ph, ea_ph, eas_ph, g):
            continue
        matched.append((p_pos, outs, p_ph, g, ea_ph, eas_ph, step_ph))
    if not matched:
        return 0

    graph = gm.graph
    out_node = next(n for n in graph.nodes if n.op == "output")

    # grad-cast folding: the autocast trace casts every bf16 grad to fp32
    # just for the optimizer (~1 activation-free but param-count-sized
    # kernel per parameter).  The HIP kernel reads bf16 grads directly —
    # unwrap the casts when EVERY matched grad unwraps uniformly.
    def _uncast_bf16(g):
        if isinstance(g, fx.Node) and g.op == "call_function"                 and g.target is aten._to_copy.default                 and g.kwargs.get("dtype") == torch.float32:
            src = g.args[0]
            v = src.meta.get("val") if hasattr(src, "meta") else None
            if isinstance(v, torch.Tensor) and v.dtype == torch.bfloat16:
                return src
        return None

    # grad-cast folding: the autocast trace casts every bf16 grad to
    # fp32 just for the optimizer. The HIP kernel reads bf16 grads
    # directly but needs a UNIFORM grad dtype per call — partition the
    # bank into a bf16-grad call and an fp32-grad call (two launches
    # instead of ~hundred cast kernels).
    unwrapped = [_uncast_bf16(m[3]) for m in matched]
    bf16_bank = [(m[0], m[1], m[2], u, m[4], m[5], m[6])
                 for m, u in zip(matched, unwrapped) if u is not None]
    fp32_bank = [m for m, u in zip(matched, unwrapped) if u is None]
    if bf16_bank:
        logger.info("fuse_optimizer: feeding %d bf16 grads directly "
                    "(folded fp32 casts; %d fp32-grad params in a second "
                    "bank)", len(bf16_bank), len(fp32_bank))
    banks = [b for b in (bf16_bank, fp32_bank) if b]

    with graph.inserting_before(out_node):
        for bank in banks:
            p_list = [m[2] for m in bank]
            if decay_scale is not None:
                # decoupled decay for the bank in one multi-tensor op
                scaled = graph.call_function(
                    aten._foreach_mul.Scalar, (p_list, decay_scale))
                p_list = [graph.call_function(operator.getitem, (scaled, i))
                          for i in range(len(bank))]
            fused = graph.call_function(
                torch.ops.easydist_amd.fused_adam_step.default,
                (p_list,                         # params (maybe pre-decayed)
                 [m[3] for m in bank],           # grads
                 [m[4] for m in bank],           # exp_avgs
                 [m[5] for m in bank],           # exp_avg_sqs
                 [m[6] for m in bank],           # steps (pre-increment)
                 lr, beta1, beta2, kernel_wd, eps))
            lists = [graph.call_function(operator.getitem, (fused, k))
                     for k in range(4)]
            for i, (p_pos, outs, *_rest) in enumerate(bank):
                items = [graph.call_function(operator.getitem,
                                             (lists[k], i))
                         for k in range(4)]
                flat_outs[outs["param"]] = items[0]
                flat_outs[outs["exp_avg"]] = items[1]
                flat_outs[outs["exp_avg_sq"]] = items[2]
                flat_outs[outs["step"]] = items[3]
    logger.info("fuse_optimizer: fused %d/%d parameter Adam chains",
                len(matched), len(param_positions))
    return len(matched)


def _match_sgd_chain(p_ph, buf_ph, p_new, buf_new, momentum, wd, nesterov):
    """Verify the decomposed momentum-SGD shape (decomp.py:
    fused_sgd_decomp, is_first_step=False); return the RAW grad node."""
    # buf_new = add(mul(buf_ph, momentum), mul(g', 1-dampening))
    if not (isinstance(buf_new, fx.Node)
            and buf_new.target is aten.add.Tensor):
        return None
    t0, s0 = _tensor_scalar(buf_new.args[0])
    t1, s1 = _tensor_scalar(buf_new.args[1])
    if t0 is buf_ph:
        g = t1
    elif t1 is buf_ph:
        g = t0
    else:
        return None
    if g is None:
        # dampening == 0 traces `1 * g` away: the addend IS g'
        g = buf_new.args[1] if t0 is buf_ph else buf_new.args[0]
    if not isinstance(g, fx.Node):
        return None
    if wd != 0.0:
        g = _strip_l2(g, p_ph, wd)
        if g is None:
            return None
    # p_new = sub(p_ph, mul(upd, lr))
    if not (isinstance(p_new, fx.Node)
            and p_new.target is aten.sub.Tensor
            and p_new.args[0] is p_ph):
        return None
    upd, _lr = _tensor_scalar(p_new.args[1])
    if upd is None:
        return None
    if nesterov:
        # upd = add(g', mul(buf_new, momentum))
        if not (isinstance(upd, fx.Node)
                and upd.target is aten.add.Tensor):
            return None
    elif upd is not buf_new:
        return None
    return g


def _fuse_sgd(gm, flat_outs, placeholders, param_positions, opt, pl_env):
    groups = opt.param_groups
    if len(groups) != 1:
        return 0
    lr = float(groups[0]["lr"])
    momentum = float(groups[0].get("momentum", 0.0))
    dampening = float(groups[0].get("dampening", 0.0))
    wd = float(groups[0].get("weight_decay", 0.0))
    nesterov = bool(groups[0].get("nesterov", False))
    if momentum == 0.0 or groups[0].get("maximize"):
        return 0     # no state to fuse / unsupported

    def _ok_placements(*nodes):
        if not pl_env:
            return True
        seen = []
        for n in nodes:
            pls = pl_env.get(n.name)
            if pls:
                seen.append(tuple(repr(p) for p in pls[0]))
        if not seen:
            return True
        if all(all(p == "R" for p in v) for v in seen):
            return True
        return len(set(seen)) == 1 and len(seen) == len(nodes)

    matched = []
    for p_pos, outs in param_positions.items():
        p_ph = placeholders[p_pos]
        buf_ph = placeholders[outs["buf_in"]]
        p_new = flat_outs[outs["param"]]
        buf_new = flat_outs[outs["buf"]]
        g = _match_sgd_chain(p_ph, buf_ph, p_new, buf_new, momentum, wd,
                             nesterov)
        if g is None:
            continue
        if not _ok_placements(p_ph, buf_ph, g):
            continue
        matched.append((p_pos, outs, p_ph, g, buf_ph))
    if not matched:
        return 0

    graph = gm.graph
    out_node = next(n for n in graph.nodes if n.op == "output")
    with graph.inserting_before(out_node):
        fused = graph.call_function(
            torch.ops.easydist_amd.fused_sgd_step.default,
            ([m[2] for m in matched], [m[3] for m in matched],
             [m[4] for m in matched], lr, momentum, dampening, wd,
             nesterov))
        lists = [graph.call_function(operator.getitem, (fused, k))
                 for k in range(2)]
        for i, (p_pos, outs, *_rest) in enumerate(matched):
            flat_outs[outs["param"]] = graph.call_function(
                operator.getitem, (lists[0], i))
            flat_outs[outs["buf"]] = graph.call_function(
                operator.getitem, (lists[1], i))
    logger.info("fuse_optimizer: fused %d/%d parameter SGD chains",
                len(matched), len(param_positions))
    return len(matched)
