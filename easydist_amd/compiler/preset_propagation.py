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

# Rules whose sharded semantics are realized by a RUNTIME exchange (ring
# attention, MoE EP) rather than by independent local execution: the
# machine-check in tests/test_presets.py skips these (op -> "all" or a
# set of input dims whose shard groups are runtime-realized).
RUNTIME_REALIZED: Dict[object, object] = {}


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
    # trailing-aligned broadcasting: an input participates in a dim's
    # shard iff it actually has that dim at full size; size-1/absent
    # (broadcast) inputs stay replicated, which is exactly correct
    out_rank = max(len(s) for s in input_shapes)
    out = []
    for o in range(out_rank):
        m = 1
        for sh in input_shapes:
            di = o - (out_rank - len(sh))
            if di >= 0 and sh[di] != 1:
                if m != 1 and sh[di] != m:
                    return None           # invalid broadcast combo
                m = sh[di]
        out.append(m)
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for o in range(out_rank):
        if out[o] <= 1:
            continue
        parts = [(i, o - (out_rank - len(sh)))
                 for i, sh in enumerate(input_shapes)
                 if o - (out_rank - len(sh)) >= 0
                 and sh[o - (out_rank - len(sh))] == out[o]]
        if not parts:
            continue
        for i, di in parts:
            ann[i][di] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(o)
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


# ---------------------------------------------------- permutation family ----
# Pure index permutations: every dim shards; only the gather position
# moves. Probing these by execution (fp64 copies of full activations)
# was the bulk of the residual GPT-2 discovery time.
def _perm_rule(input_shapes, perm_of_out):
    """perm_of_out[out_dim] = in_dim."""
    shape = input_shapes[0]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for o, i in enumerate(perm_of_out):
        if shape[i] <= 1:
            continue
        ann[0][i] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(o)
        sid += 1
    return ann, combs


@register_preset(aten.t.default)
def _t_rule(input_shapes, args, kwargs):
    if len(input_shapes[0]) != 2:
        return None
    return _perm_rule(input_shapes, [1, 0])


@register_preset(aten.transpose.int)
def _transpose_rule(input_shapes, args, kwargs):
    r = len(input_shapes[0])
    d0, d1 = args[1] % r, args[2] % r
    perm = list(range(r))
    perm[d0], perm[d1] = perm[d1], perm[d0]
    return _perm_rule(input_shapes, perm)


@register_preset(aten.permute.default)
def _permute_rule(input_shapes, args, kwargs):
    r = len(input_shapes[0])
    return _perm_rule(input_shapes, [d % r for d in args[1]])


@register_preset(aten.unsqueeze.default)
def _unsqueeze_rule(input_shapes, args, kwargs):
    shape = input_shapes[0]
    r = len(shape)
    d = args[1] % (r + 1)
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for i in range(r):
        if shape[i] <= 1:
            continue
        ann[0][i] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(i + (1 if i >= d else 0))
        sid += 1
    return ann, combs


@register_preset(aten.squeeze.dim, aten.squeeze.dims, aten.squeeze.default)
def _squeeze_rule(input_shapes, args, kwargs):
    shape = input_shapes[0]
    r = len(shape)
    if len(args) < 2:
        drop = [i for i in range(r) if shape[i] == 1]
    else:
        ds = args[1] if isinstance(args[1], (list, tuple)) else [args[1]]
        drop = [d % r for d in ds if shape[d % r] == 1]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for i in range(r):
        if shape[i] <= 1 or i in drop:
            continue
        o = i - sum(1 for d in drop if d < i)
        ann[0][i] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(o)
        sid += 1
    return ann, combs


# --------------------------------------------------------- reductions -------
@register_preset(aten.sum.dim_IntList, aten.mean.dim)
def _dim_reduce_rule(input_shapes, args, kwargs):
    shape = input_shapes[0]
    r = len(shape)
    dims = args[1] if len(args) > 1 and args[1] is not None \
        else list(range(r))
    dims = [d % r for d in (dims if isinstance(dims, (list, tuple))
                            else [dims])]
    keep = bool(args[2]) if len(args) > 2 else bool(kwargs.get("keepdim"))
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for i in range(r):
        if shape[i] <= 1:
            continue
        if i in dims:
            continue   # reduced-dim shard would be PARTIAL(sum) for sum /
                       # not representable for mean; conservatively omitted
        o = i if keep else i - sum(1 for d in dims if d < i)
        ann[0][i] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(o)
        sid += 1
    return ann, combs


@register_preset(aten.pow.Scalar)
def _pow_scalar_rule(input_shapes, args, kwargs):
    # pow.Scalar(scalar_base, tensor_exponent): elementwise in the tensor
    shape = input_shapes[0]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(shape)):
        if shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    if not combs:
        return None
    return ann, combs


# ------------------------------------------------------------ cat/split -----
@register_preset(aten.cat.default)
def _cat_rule(input_shapes, args, kwargs):
    if not input_shapes:
        return None
    r = len(input_shapes[0])
    cdim = (args[1] if len(args) > 1 else kwargs.get("dim", 0)) % r
    if any(len(s) != r for s in input_shapes):
        return None
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(r):
        if d == cdim:
            continue
        if any(s[d] != input_shapes[0][d] for s in input_shapes) \
                or input_shapes[0][d] <= 1:
            continue
        for i in range(len(input_shapes)):
            ann[i][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


@register_preset(aten.split.Tensor)
def _split_rule(input_shapes, args, kwargs):
    import math as _math
    shape = input_shapes[0]
    r = len(shape)
    sdim = (args[2] if len(args) > 2 else kwargs.get("dim", 0)) % r
    size = args[1]
    if isinstance(size, (list, tuple)):
        n_out = len(size)
    else:
        n_out = _math.ceil(shape[sdim] / size) if size else 0
    if n_out <= 0:
        return None
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(r):
        if d == sdim or shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_gather(d)] * n_out
        sid += 1
    return ann, combs


# ------------------------------------------------------------ layer norm ----
@register_preset(aten.native_layer_norm.default)
def _ln_rule(input_shapes, args, kwargs):
    # tensors: (x[, w][, b]); normalized = trailing len(args[1]) dims
    x = input_shapes[0]
    n_norm = len(args[1])
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(x) - n_norm):
        if x[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_gather(d), _gather(d), _gather(d)]
        sid += 1
    return ann, combs


@register_preset(aten.native_layer_norm_backward.default)
def _ln_bwd_rule(input_shapes, args, kwargs):
    # tensors: (gout, x, mean, rstd[, w][, b]); outputs (dx, dw, db);
    # batch-dim shard -> dx gathers, dw/db are PARTIAL sums
    x = input_shapes[1]
    n_norm = len(args[2])
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(len(x) - n_norm):
        if x[d] <= 1:
            continue
        for i in range(4):      # gout, x, mean, rstd share batch layout
            if d < len(input_shapes[i]):
                ann[i][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_gather(d), _reduce_add(), _reduce_add()]
        sid += 1
    return ann, combs


# ------------------------------------------------- easydist_amd custom ops --
def _register_custom_ops():
    """Sharding algebra of our own gfx950 kernels is known exactly; their
    fp64 aten fallbacks are among the most expensive things execution
    discovery can run (full-activation flash attention, [N, vocab] CE)."""
    try:
        from .. import ops as _ops  # noqa: F401  (defines torch.ops.easydist_amd)
    except Exception:               # pragma: no cover
        return
    ed = torch.ops.easydist_amd

    @register_preset(ed.flash_attention.default)
    def _fa_rule(input_shapes, args, kwargs):
        # (q,k,v [B,H,S,D]) -> (out [B,H,S,D], lse [B,H,S])
        if len(input_shapes[0]) != 4:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        combs = {}
        sid = 1
        for d in (0, 1):
            if input_shapes[0][d] <= 1:
                continue
            for i in range(3):
                ann[i][d] = ShardDim.get_shard_dim(sid)
            combs[sid] = [_gather(d), _gather(d)]
            sid += 1
        # sequence-parallel strategy: S(2) on q/k/v is realized by the
        # ring-attention runtime (the sharding transform rewrites the op
        # to rt_ring_attention) — long-context choice for the solver
        RUNTIME_REALIZED[ed.flash_attention.default] = {2}
        if input_shapes[0][2] > 1:
            for i in range(3):
                ann[i][2] = ShardDim.get_shard_dim(sid)
            combs[sid] = [_gather(2), _gather(2)]
        return ann, combs

    @register_preset(ed.flash_attention_bwd.default)
    def _fa_bwd_rule(input_shapes, args, kwargs):
        # (grad,q,k,v,out [B,H,S,D], lse [B,H,S]) -> (dq,dk,dv)
        if len(input_shapes) < 6 or len(input_shapes[1]) != 4:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        combs = {}
        sid = 1
        for d in (0, 1):
            if input_shapes[1][d] <= 1:
                continue
            for i in range(6):
                ann[i][d] = ShardDim.get_shard_dim(sid)
            combs[sid] = [_gather(d)] * 3
            sid += 1
        # sequence-parallel: rewritten to rt_ring_attention_bwd
        RUNTIME_REALIZED[ed.flash_attention_bwd.default] = {2}
        if input_shapes[1][2] > 1:
            for i in range(6):
                ann[i][2] = ShardDim.get_shard_dim(sid)
            combs[sid] = [_gather(2)] * 3
        return ann, combs

    @register_preset(ed.ce_fwd.default)
    def _ce_rule(input_shapes, args, kwargs):
        # (logits [N,C], targets [N]) -> (nll SUM scalar, lse [N]):
        # row shard -> loss PARTIAL(sum), lse gathers; class shard not
        # representable (partial logsumexp)
        if len(input_shapes[0]) != 2 or input_shapes[0][0] <= 1:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        ann[0][0] = ShardDim.get_shard_dim(1)
        ann[1][0] = ShardDim.get_shard_dim(1)
        return ann, {1: [_reduce_add(), _gather(0)]}

    # ---- MoE routing/combine (EP semantics -- see ops/moe_ops.py) ----
    for _op in (ed.moe_bins.default, ed.moe_bins_bwd.default,
                ed.moe_combine.default, ed.moe_combine_bwd.default):
        RUNTIME_REALIZED[_op] = "all"
    @register_preset(ed.moe_bins.default)
    def _moe_bins_rule(input_shapes, args, kwargs):
        # (tokens [N,C], topi [N,K], topv [N,K]) -> bins/gates/src/valid
        # [E,cap,*]; token shard -> capacity-slice shard (parallel routing)
        if len(input_shapes[0]) != 2 or input_shapes[0][0] <= 1:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        for i in range(3):
            ann[i][0] = ShardDim.get_shard_dim(1)
        return ann, {1: [_gather(1), _gather(1), _gather(1), _gather(1)]}

    @register_preset(ed.moe_bins_bwd.default)
    def _moe_bins_bwd_rule(input_shapes, args, kwargs):
        # (gbins [E,cap,C], ggates [E,cap], src, valid, tokens_ref [N,C])
        # -> (gtokens [N,C], gtopv [N,K]); cap-slice shard -> token shard
        if len(input_shapes[0]) != 3 or input_shapes[0][1] <= 1:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        for i in range(4):
            ann[i][1] = ShardDim.get_shard_dim(1)
        ann[4][0] = ShardDim.get_shard_dim(1)
        return ann, {1: [_gather(0), _gather(0)]}

    @register_preset(ed.moe_combine.default)
    def _moe_combine_rule(input_shapes, args, kwargs):
        # (bins [E,cap,C], gates, src, valid, tokens_ref) -> out [N,C]
        if len(input_shapes[0]) != 3 or input_shapes[0][1] <= 1:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        for i in range(4):
            ann[i][1] = ShardDim.get_shard_dim(1)
        ann[4][0] = ShardDim.get_shard_dim(1)
        return ann, {1: _gather(0)}

    @register_preset(ed.moe_combine_bwd.default)
    def _moe_combine_bwd_rule(input_shapes, args, kwargs):
        # (gout [N,C], bins [E,cap,C], gates, src, valid) -> (gbins, ggates)
        if len(input_shapes[1]) != 3 or input_shapes[1][1] <= 1:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        ann[0][0] = ShardDim.get_shard_dim(1)
        for i in range(1, 5):
            ann[i][1] = ShardDim.get_shard_dim(1)
        return ann, {1: [_gather(1), _gather(1)]}

    @register_preset(ed.ce_bwd.default)
    def _ce_bwd_rule(input_shapes, args, kwargs):
        # (grad scalar, logits [N,C], targets [N], lse [N]) -> dlogits
        if len(input_shapes) < 4 or len(input_shapes[1]) != 2 \
                or input_shapes[1][0] <= 1:
            return None, {}
        ann = ShardAnnotation.init_from_input_shapes(input_shapes)
        for i in (1, 2, 3):
            ann[i][0] = ShardDim.get_shard_dim(1)
        return ann, {1: _gather(0)}


_register_custom_ops()


# ------------------------------------------------------------ conv / bn -----
# fp64 execution probes of full-size convolutions are the dominant
# discovery cost on the (w)resnet family; the algebra is closed-form.
@register_preset(aten.convolution.default)
def _conv_rule(input_shapes, args, kwargs):
    # (x [N,Ci,*sp], w [Co,Ci/g,*k][, bias [Co]]); groups = args[-1]
    if len(input_shapes) < 2 or len(input_shapes[0]) < 3:
        return None
    x, w = input_shapes[0], input_shapes[1]
    has_bias = len(input_shapes) > 2
    transposed = bool(args[6]) if len(args) > 6 else False
    groups = args[8] if len(args) > 8 else 1
    if transposed:
        return None
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    if x[0] > 1:                       # batch
        ann[0][0] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(0)
        sid += 1
    if groups == 1 and w[0] > 1:       # out channels: w + bias
        ann[1][0] = ShardDim.get_shard_dim(sid)
        if has_bias:
            ann[2][0] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(1)
        sid += 1
    if groups == 1 and not has_bias and x[1] > 1:   # in channels: PARTIAL
        ann[0][1] = ShardDim.get_shard_dim(sid)
        ann[1][1] = ShardDim.get_shard_dim(sid)
        combs[sid] = _reduce_add()
        sid += 1
    return ann, combs


@register_preset(aten.convolution_backward.default)
def _conv_bwd_rule(input_shapes, args, kwargs):
    # (gout [N,Co,*], x [N,Ci,*], w [Co,Ci/g,*k]) -> (dx, dw, db)
    if len(input_shapes) < 3 or len(input_shapes[0]) < 3:
        return None
    transposed = bool(args[7]) if len(args) > 7 else False
    groups = args[9] if len(args) > 9 else 1
    if transposed:
        return None
    gout, x, w = input_shapes[:3]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    if x[0] > 1:                       # batch: dw/db are partial sums
        ann[0][0] = ShardDim.get_shard_dim(sid)
        ann[1][0] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_gather(0), _reduce_add(), _reduce_add()]
        sid += 1
    if groups == 1 and gout[1] > 1:    # out channels: dx partial
        ann[0][1] = ShardDim.get_shard_dim(sid)
        ann[2][0] = ShardDim.get_shard_dim(sid)
        combs[sid] = [_reduce_add(), _gather(0), _gather(0)]
        sid += 1
    return ann, combs


def _bn_rule_n(input_shapes, n_out):
    # channel shard only: batch/spatial sharding changes the TRAINING
    # batch statistics (per-shard mean != global mean), so it must not
    # be offered. tensors: (x [N,C,*], w [C], b [C][, rm [C], rv [C]])
    x = input_shapes[0]
    if len(x) < 2 or x[1] <= 1:
        return None, {}
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    ann[0][1] = ShardDim.get_shard_dim(1)
    for i in range(1, len(input_shapes)):
        if len(input_shapes[i]) == 1 and input_shapes[i][0] == x[1]:
            ann[i][0] = ShardDim.get_shard_dim(1)
    # out gathers on channel; save_mean/save_var (and the functional
    # variant's updated running stats) gather on their dim 0
    combs = {1: [_gather(1)] + [_gather(0)] * (n_out - 1)}
    return ann, combs


@register_preset(aten.native_batch_norm.default,
                 aten._native_batch_norm_legit.default)
def _bn_rule(input_shapes, args, kwargs):
    return _bn_rule_n(input_shapes, 3)


@register_preset(aten._native_batch_norm_legit_functional.default)
def _bn_func_rule(input_shapes, args, kwargs):
    return _bn_rule_n(input_shapes, 5)


@register_preset(aten.native_batch_norm_backward.default)
def _bn_bwd_rule(input_shapes, args, kwargs):
    # (gout, x, w, rm, rv, sm, sv) -> (dx, dw, db): channel shard
    x = input_shapes[1] if len(input_shapes) > 1 else None
    if x is None or len(x) < 2 or x[1] <= 1:
        return None, {}
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    ann[0][1] = ShardDim.get_shard_dim(1)
    ann[1][1] = ShardDim.get_shard_dim(1)
    for i in range(2, len(input_shapes)):
        if len(input_shapes[i]) == 1 and input_shapes[i][0] == x[1]:
            ann[i][0] = ShardDim.get_shard_dim(1)
    return ann, {1: [_gather(1), _gather(0), _gather(0)]}


@register_preset(aten.max_pool2d_with_indices.default)
def _maxpool_rule(input_shapes, args, kwargs):
    # indices are within-plane offsets: invariant under N/C sharding
    x = input_shapes[0]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in (0, 1):
        if d < len(x) - 2 and x[d] > 1:
            ann[0][d] = ShardDim.get_shard_dim(sid)
            combs[sid] = [_gather(d), _gather(d)]
            sid += 1
    return ann, combs


@register_preset(aten.max_pool2d_with_indices_backward.default)
def _maxpool_bwd_rule(input_shapes, args, kwargs):
    # (gout, x, ..., indices) -> dx
    x = input_shapes[1] if len(input_shapes) > 1 else None
    if x is None:
        return None
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in (0, 1):
        if d < len(x) - 2 and x[d] > 1:
            for i in range(len(input_shapes)):
                if d < len(input_shapes[i]) - 2:
                    ann[i][d] = ShardDim.get_shard_dim(sid)
            combs[sid] = _gather(d)
            sid += 1
    return ann, combs


# ----------------------------------------------------------- softmax --------
@register_preset(aten._softmax.default, aten._log_softmax.default,
                 aten.softmax.int, aten.log_softmax.int)
def _softmax_rule(input_shapes, args, kwargs):
    # (x, dim, ...): every dim except the normalized one shards
    shape = input_shapes[0]
    r = len(shape)
    sdim = args[1] % r if shape else 0
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(r):
        if d == sdim or shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


@register_preset(aten._softmax_backward_data.default,
                 aten._log_softmax_backward_data.default)
def _softmax_bwd_rule(input_shapes, args, kwargs):
    # (gout, out, dim, dtype)
    shape = input_shapes[0]
    r = len(shape)
    sdim = args[2] % r if shape else 0
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(r):
        if d == sdim or shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        ann[1][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


@register_preset(aten.tril.default, aten.triu.default)
def _tri_rule(input_shapes, args, kwargs):
    # triangular masks act on the LAST TWO dims; batch dims shard
    shape = input_shapes[0]
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(max(0, len(shape) - 2)):
        if shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs


@register_preset(aten.constant_pad_nd.default)
def _pad_rule(input_shapes, args, kwargs):
    # pad spec covers trailing dims (reversed pairs); unpadded dims shard
    shape = input_shapes[0]
    r = len(shape)
    pad = list(args[1])
    padded = set()
    for i in range(len(pad) // 2):
        if pad[2 * i] or pad[2 * i + 1]:
            padded.add(r - 1 - i)
    ann = ShardAnnotation.init_from_input_shapes(input_shapes)
    combs = {}
    sid = 1
    for d in range(r):
        if d in padded or shape[d] <= 1:
            continue
        ann[0][d] = ShardDim.get_shard_dim(sid)
        combs[sid] = _gather(d)
        sid += 1
    return ann, combs
