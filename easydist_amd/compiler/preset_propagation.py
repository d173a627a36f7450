"""Preset sharding rules for ops where execution-based discovery is wrong
or wasteful (randomness, views, factories).

Capability parity with reference ``easydist/torch/preset_propagation.py``
(registry + rules for placeholder/view/factory/dropout ops, lines 28-130).
"""
from __future__ import annotations

import functools
from typing import Callable, Dict, List, Optional, Tuple

import torch

from ..metashard.annotation import NoShardDim, ShardAnnotation, ShardDim
from ..metashard.combination import CombinationFunc
from ..metashard.view_propagation import view_propagation

aten = torch.ops.aten

# op overload -> fn(input_shapes, args, kwargs) -> (ShardAnnotation, {id: comb})
_PRESET_REGISTRY: Dict[object, Callable] = {}


def register_preset(*ops):
    def deco(fn):
        for op in ops:
            _PRESET_REGISTRY[op] = fn
        return fn
    return deco


def preset_meta_spmd(op, input_shapes, args, kwargs):
    fn = _PRESET_REGISTRY.get(op)
    if fn is None:
        return None
    return fn(input_shapes, args, kwargs)


def _gather(d):
    return functools.partial(CombinationFunc.gather, dim=d)


# ------------------------------------------------------------------ views ----
@register_preset(aten.view.default, aten._unsafe_view.default,
                 aten.reshape.default)
def _view_rule(input_shapes, args, kwargs):
    in_shape = list(input_shapes[0])
    out_shape = list(args[1])
    mapping = view_propagation(in_shape, out_shape)
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for in_dim, out_dim in mapping.items():
        if in_shape[in_dim] <= 1:
            continue
        ann[0][in_dim] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(out_dim)
        sid += 1
    return ann, combs


@register_preset(aten.expand.default)
def _expand_rule(input_shapes, args, kwargs):
    in_shape = list(input_shapes[0])
    out_shape = list(args[1])
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    # trailing-aligned dims; an input dim only shards when it is not being
    # broadcast (size preserved and > 1)
    off = len(out_shape) - len(in_shape)
    for i, s in enumerate(in_shape):
        o = out_shape[i + off]
        if s > 1 and (o == s or o == -1):
            ann[0][i] = ShardDim.get_shard_dim(sid)
            combs[sid] = _gather(i + off)
            sid += 1
    return ann, combs


# ---------------------------------------------------------------- dropout ----
@register_preset(aten.native_dropout.default)
def _dropout_rule(input_shapes, args, kwargs):
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(input_shapes[0])):
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_gather(d), _gather(d)]   # output + mask
        sid += 1
    return ann, combs


@register_preset(aten.native_dropout_backward.default)
def _dropout_bwd_rule(input_shapes, args, kwargs):
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(input_shapes[0])):
        ann[0][d] = ShardDim.get_shard_dim(sid)
        if len(input_shapes) > 1 and d < len(input_shapes[1]):
            ann[1][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


# -------------------------------------------------------------- factories ----
@register_preset(aten.ones_like.default, aten.zeros_like.default,
                 aten.empty_like.default, aten.full_like.default,
                 aten.rand_like.default, aten.randn_like.default,
                 aten.clone.default, aten.detach.default,
                 aten._to_copy.default, aten.alias.default,
                 aten.contiguous.default)
def _like_rule(input_shapes, args, kwargs):
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(input_shapes[0])):
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


# -------------------------------------------------------------- embedding ----
@register_preset(aten.embedding.default)
def _embedding_rule(input_shapes, args, kwargs):
    # inputs: (weight [V, D], indices [...])
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    # weight embedding-dim shard -> output last dim shard
    if len(input_shapes[0]) == 2:
        ann[0][1] = ShardDim.get_shard_dim(sid)
        out_rank = len(input_shapes[1]) + 1
        combs[sid] = _gather(out_rank - 1)
        sid += 1
    # indices dims shard -> output same dim shard
    for d in range(len(input_shapes[1])):
        ann[1][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


@register_preset(aten.embedding_dense_backward.default)
def _embedding_bwd_rule(input_shapes, args, kwargs):
    # inputs: (grad_out [..., D], indices [...]); output [V, D]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    # shard grad_out feature dim -> output dim 1 shard
    g_rank = len(input_shapes[0])
    ann[0][g_rank - 1] = ShardDim.get_shard_dim(sid)
    combs[sid] = _gather(1)
    sid += 1
    # shard batch dims of grad+indices together -> PARTIAL(sum) grad weight
    for d in range(len(input_shapes[1])):
        ann[0][d] = ShardDim.get_shard_dim(sid)
        ann[1][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = functools.partial(CombinationFunc.reduce,
                                       ops=__import__("operator").add)
        sid += 1
    return ann, combs


# ------------------------------------------------------- slice family -------
# slice/narrow/select keep a fixed window of ONE dim: every OTHER dim
# shards freely; the windowed dim must stay whole (a shard's local window
# reads the wrong global elements — discovered the hard way when all-zero
# integer probes made a dim-sharded slice of position_ids verify as
# identity).
@register_preset(aten.slice.Tensor)
def _slice_rule(input_shapes, args, kwargs):
    shape = input_shapes[0]
    sdim = args[1] if len(args) > 1 else 0
    sdim = sdim % len(shape) if shape else 0
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(shape)):
        if d == sdim or shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


@register_preset(aten.narrow.default)
def _narrow_rule(input_shapes, args, kwargs):
    return _slice_rule(input_shapes, args, kwargs)


@register_preset(aten.select.int)
def _select_rule(input_shapes, args, kwargs):
    shape = input_shapes[0]
    sdim = args[1] % len(shape) if shape else 0
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(shape)):
        if d == sdim or shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d - (1 if d > sdim else 0))
        sid += 1
    return ann, combs


# ------------------------------------------------------------- attention ----
# Fused SDPA kernels must NEVER be probed by execution: the CPU flash
# kernel SIGFPEs (integer div-by-zero, uncatchable) on some sharded
# probe combos, and the algebra is closed-form anyway: batch and head
# dims shard freely (attention mixes only within a head's sequence),
# an attn_mask shards along iff its dim matches (1 = broadcast stays
# replicated).
def _sdpa_shardable_dims(shapes, n_qkv, mask_idx):
    """Yield (dim, participating-input-indices) for batch/head dims."""
    q = shapes[0]   # bwd's grad_out has q's [B,H,S,D] layout too
    for dim in (0, 1):
        size = q[dim]
        if size <= 1:
            continue
        idxs = list(range(n_qkv))
        if mask_idx is not None:
            m = shapes[mask_idx]
            if len(m) == 4 and m[dim] == size:
                idxs.append(mask_idx)
            elif len(m) == 4 and m[dim] != 1:
                continue                          # incompatible mask
        yield dim, idxs


@register_preset(
    aten._scaled_dot_product_flash_attention_for_cpu.default)
def _sdpa_cpu_rule(input_shapes, args, kwargs):
    # tensors: (q, k, v[, attn_mask]); outputs (out [B,H,S,D], lse [B,H,S])
    if len(input_shapes[0]) != 4:
        return None, {}
    mask_idx = 3 if len(input_shapes) > 3 \
        and len(input_shapes[3]) == 4 else None
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for dim, idxs in _sdpa_shardable_dims(input_shapes, 3, mask_idx):
        for i in idxs:
            ann[i][dim] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_gather(dim), _gather(dim)]
        sid += 1
    return ann, combs


@register_preset(
    aten._scaled_dot_product_flash_attention_for_cpu_backward.default)
def _sdpa_cpu_bwd_rule(input_shapes, args, kwargs):
    # tensors: (grad, q, k, v, out, lse[, attn_mask]); outputs (dq,dk,dv)
    if len(input_shapes) < 6 or len(input_shapes[1]) != 4:
        return None, {}
    mask_idx = 6 if len(input_shapes) > 6 \
        and len(input_shapes[6]) == 4 else None
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for dim, idxs in _sdpa_shardable_dims(input_shapes, 6, mask_idx):
        for i in idxs:
            ann[i][dim] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_gather(dim)] * 3
        sid += 1
    return ann, combs


# ------------------------------------------------------- pipeline markers ----
def _register_pp_markers():
    """pp_split/step_split are identities: shard through any dim.

    Registered lazily because the marker ops live in parallel.pp.split
    (imported on first pipeline compile)."""
    try:
        import easydist_amd.parallel.pp.split  # noqa: F401
        ops = [torch.ops.easydist_amd.pp_split.default,
               torch.ops.easydist_amd.step_split.default]
    except (ImportError, AttributeError):
        return

    @register_preset(*ops)
    def _marker_rule(input_shapes, args, kwargs):
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        combs = {}
        sid = 1
        for d in range(len(input_shapes[0])):
            ann[0][d] = ShardDim.get_shard_dim(sid)
            combs[sid] = _gather(d)
            sid += 1
        return ann, combs


_register_pp_markers()


# ---------------------------------------------------------- matmul family ----
# Analytic rules: execution-based discovery on the matmul family costs the
# bulk of compile time (fp64 probes of [*,14336,4096] bmms on MoE graphs);
# the algebra is closed-form. reference kept these in the DFS; we preset.
def _reduce_add():
    import operator as _operator
    return functools.partial(CombinationFunc.reduce, ops=_operator.add)


@register_preset(aten.mm.default)
def _mm_rule(input_shapes, args, kwargs):
    (M, K), (K2, N) = input_shapes[0], input_shapes[1]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    ann[0][0] = ShardDim.get_shard_dim(1)          # M
    combs[1] = _gather(0)
    ann[1][1] = ShardDim.get_shard_dim(2)          # N
    combs[2] = _gather(1)
    ann[0][1] = ShardDim.get_shard_dim(3)          # K (both operands)
    ann[1][0] = ShardDim.get_shard_dim(3)
    combs[3] = _reduce_add()
    return ann, combs


@register_preset(aten.addmm.default)
def _addmm_rule(input_shapes, args, kwargs):
    # (bias, a[M,K], b[K,N])
    bias = input_shapes[0]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    ann[1][0] = ShardDim.get_shard_dim(1)          # M
    if len(bias) == 2 and bias[0] == input_shapes[1][0]:
        ann[0][0] = ShardDim.get_shard_dim(1)
    combs[1] = _gather(0)
    ann[2][1] = ShardDim.get_shard_dim(2)          # N
    if len(bias) >= 1 and bias[-1] == input_shapes[2][1]:
        ann[0][len(bias) - 1] = ShardDim.get_shard_dim(2)
    combs[2] = _gather(1)
    # K-shard omitted: PARTIAL would double-count the bias term
    return ann, combs


@register_preset(aten.bmm.default)
def _bmm_rule(input_shapes, args, kwargs):
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    ann[0][0] = ShardDim.get_shard_dim(1)          # batch
    ann[1][0] = ShardDim.get_shard_dim(1)
    combs[1] = _gather(0)
    ann[0][1] = ShardDim.get_shard_dim(2)          # M
    combs[2] = _gather(1)
    ann[1][2] = ShardDim.get_shard_dim(3)          # N
    combs[3] = _gather(2)
    ann[0][2] = ShardDim.get_shard_dim(4)          # K
    ann[1][1] = ShardDim.get_shard_dim(4)
    combs[4] = _reduce_add()
    return ann, combs


# ------------------------------------------------------------- elementwise ---
# Same-shape elementwise ops: every dim shards with gather recombination.
# Probing these by execution is pure waste (they dominate op counts in
# decomposed optimizer + MoE graphs).
_EW_UNARY = [
    aten.neg.default, aten.sqrt.default, aten.rsqrt.default,
    aten.exp.default, aten.log.default, aten.relu.default,
    aten.gelu.default, aten.silu.default, aten.tanh.default,
    aten.sigmoid.default, aten.reciprocal.default, aten.abs.default,
    aten.sin.default, aten.cos.default, aten.erf.default,
    aten.logical_not.default, aten.sgn.default, aten.sign.default,
    aten.floor.default, aten.ceil.default, aten.round.default,
    aten.pow.Tensor_Scalar, aten.clamp.default, aten.clamp_min.default,
    aten.clamp_max.default, aten.leaky_relu.default, aten.elu.default,
    aten.hardtanh.default, aten.gelu_backward.default,
    aten.threshold_backward.default, aten.tanh_backward.default,
    aten.sigmoid_backward.default, aten.silu_backward.default,
    aten.leaky_relu_backward.default, aten.elu_backward.default,
]
_EW_BINARY = [
    aten.add.Tensor, aten.sub.Tensor, aten.mul.Tensor, aten.div.Tensor,
    aten.maximum.default, aten.minimum.default, aten.fmod.Tensor,
    aten.remainder.Tensor, aten.atan2.default, aten.pow.Tensor_Tensor,
    aten.add.Scalar, aten.sub.Scalar, aten.mul.Scalar, aten.div.Scalar,
    aten.rsub.Scalar, aten.fmod.Scalar,
    aten.eq.Tensor, aten.ne.Tensor, aten.lt.Tensor, aten.le.Tensor,
    aten.gt.Tensor, aten.ge.Tensor, aten.eq.Scalar, aten.ne.Scalar,
    aten.lt.Scalar, aten.le.Scalar, aten.gt.Scalar, aten.ge.Scalar,
    aten.logical_and.default, aten.logical_or.default,
    aten.bitwise_and.Tensor, aten.bitwise_or.Tensor,
    aten.addcmul.default, aten.addcdiv.default, aten.lerp.Scalar,
    aten.lerp.Tensor, aten.where.self,
]


@register_preset(*(_EW_UNARY + _EW_BINARY))
def _elementwise_rule(input_shapes, args, kwargs):
    if not input_shapes:
        return None
    shape0 = input_shapes[0]
    # every TENSOR input must have the exact same shape (no broadcasting:
    # broadcast patterns go to execution-based discovery)
    for sh in input_shapes[1:]:
        if tuple(sh) != tuple(shape0):
            return None
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(shape0)):
        if shape0[d] <= 1:
            continue
        for i in range(len(input_shapes)):
            ann[i][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    if not combs:
        return None
    return ann, combs


# -------------------------------------------------- value-dependent ops ------
# Ops whose semantics depend on tensor VALUES (indices/orderings): probe
# executions use zero-filled integer tensors (OOB-safety), which makes
# sharded-recombination checks pass COINCIDENTALLY and mints false rules.
# Force replicate: (None, {}) = "no sharding rule".
_VALUE_DEPENDENT = [
    aten.index.Tensor, aten.index_put.default, aten.index_add.default,
    aten.index_select.default, aten.index_copy.default,
    aten.scatter.src, aten.scatter.value, aten.scatter_add.default,
    aten.scatter_reduce.two, aten.gather.default,
    aten.argsort.default, aten.sort.default, aten.topk.default,
    aten.searchsorted.Tensor, aten.searchsorted.Scalar,
    aten.unique_consecutive.default, aten.nonzero.default,
    aten.masked_select.default, aten.take.default, aten.bucketize.Tensor,
    aten.argmax.default, aten.argmin.default, aten.bincount.default,
    aten.mode.default, aten.kthvalue.default, aten.median.default,
    aten.index_put_.default,
]


@register_preset(*_VALUE_DEPENDENT)
def _value_dependent_rule(input_shapes, args, kwargs):
    return (None, {})
