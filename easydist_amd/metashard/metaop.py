"""MetaOp: execution-based SPMD rule discovery (ShardCombine).

Capability parity with the reference's ``easydist/metashard/metaop.py``
(MetaOp.exec / sharding_discovery, reference lines 60-260) re-designed:
instead of the reference's annotation-space DFS we search shard *groups*
greedily — for each not-yet-annotated (input, dim) seed we try sharding it
alone, then try pairing it with one dim of each later input (covers
contraction dims and broadcast-elementwise groups), then greedily extend the
group with any remaining compatible dim. Each successful group gets a fresh
shard_dim_id and its combination function recorded.
"""
from __future__ import annotations

import logging
from typing import Callable, Dict, List, Optional, Tuple

import torch

from .. import config as mdconfig
from .annotation import NoShardDim, ShardAnnotation, ShardDim
from .combination import HaloHint, try_combination

logger = logging.getLogger(__name__)


def shard_tensor(tensor: torch.Tensor, dim: int, num_shards: int,
                 halo: int = 0, chunk: int = 1) -> List[torch.Tensor]:
    """Split `tensor` along `dim` into `num_shards` pieces.

    halo > 0 widens each piece by `halo` elements toward each neighbour
    (stencil ops); chunk > 1 deals block-cyclic pieces.
    """
    size = tensor.shape[dim]
    if chunk > 1:
        blocks = torch.chunk(tensor, num_shards * chunk, dim=dim)
        shards = []
        for i in range(num_shards):
            mine = [blocks[c * num_shards + i] for c in range(chunk)]
            shards.append(torch.concat(mine, dim=dim))
        return shards
    base = size // num_shards
    bounds = [0]
    rem = size - base * num_shards
    for i in range(num_shards):
        bounds.append(bounds[-1] + base + (1 if i < rem else 0))
    shards = []
    for i in range(num_shards):
        lo = max(bounds[i] - (halo if i > 0 else 0), 0)
        hi = min(bounds[i + 1] + (halo if i < num_shards - 1 else 0), size)
        idx = [slice(None)] * tensor.dim()
        idx[dim] = slice(lo, hi)
        # contiguous: some aten CPU kernels (native_layer_norm_backward)
        # silently misread strided views — a view shard makes discovery
        # verification compare garbage numerics
        shards.append(tensor[tuple(idx)].contiguous())
    return shards


class MetaOp:
    """Wrap an op + concrete args for execution-based rule discovery."""

    def __init__(self, func: Callable, input_args: Tuple, kwargs: Optional[dict] = None,
                 name: str = ""):
        import torch.utils._pytree as pytree
        self.func = func
        self.kwargs = kwargs or {}
        self.name = name or getattr(func, "__name__", str(func))
        # stencil ops (conv/pool family): on a plain failure retry the group
        # with halo-widened input shards (boundary overlap)
        self.try_halo = any(k in self.name for k in ("conv", "pool"))
        # flat list of tensor inputs (what the annotation indexes) — pytree
        # flattened so list-typed args (cat, stack, foreach) participate
        self._flat_args, self._args_spec = pytree.tree_flatten(
            (input_args, self.kwargs))
        self.flat_tensors: List[torch.Tensor] = [
            a for a in self._flat_args if isinstance(a, torch.Tensor)
        ]

    def _call_with_tensors(self, tensors: List[torch.Tensor]):
        import torch.utils._pytree as pytree
        it = iter(tensors)
        flat = [next(it) if isinstance(a, torch.Tensor) else a
                for a in self._flat_args]
        args, kwargs = pytree.tree_unflatten(flat, self._args_spec)
        return self.func(*args, **kwargs)

    def exec_global(self):
        return self._call_with_tensors(self.flat_tensors)

    def exec_sharded(self, annotation: ShardAnnotation, shard_dim_id: int,
                     num_shards: int):
        """Run the op once per shard, sharding every dim tagged shard_dim_id."""
        per_shard_inputs: List[List[torch.Tensor]] = [[] for _ in range(num_shards)]
        for i, t in enumerate(self.flat_tensors):
            tagged = [(d, sd) for d, sd in enumerate(annotation[i])
                      if sd.shard_dim_id == shard_dim_id]
            if not tagged:
                for s in range(num_shards):
                    per_shard_inputs[s].append(t)
                continue
            if len(tagged) > 1:
                raise ValueError("one shard_dim_id twice in one input")
            d, sd = tagged[0]
            shards = shard_tensor(t, d, num_shards, halo=sd.halo, chunk=sd.chunk)
            for s in range(num_shards):
                per_shard_inputs[s].append(shards[s])
        # meta dry-run first: many aten CUDA kernels skip shape validation
        # and OOB-fault on inconsistent shard combinations; the meta kernels
        # DO validate, so a malformed combo dies here on the CPU instead of
        # taking down the GPU.
        try:
            metas = [t.to("meta") for t in per_shard_inputs[0]]
            self._call_with_tensors(metas)
        except NotImplementedError:
            pass   # no meta kernel: proceed (plain eager will raise cleanly)
        outs = []
        for s in range(num_shards):
            outs.append(self._call_with_tensors(per_shard_inputs[s]))
        return outs

    # ------------------------------------------------------------ discovery --
    def _group_works(self, annotation: ShardAnnotation, shard_dim_id: int,
                     num_shards: int, global_out, allow_halo: bool = True):
        """Try the tagged group; returns the combination or None."""
        comb = None
        halo_hinted = False
        try:
            sharded = self.exec_sharded(annotation, shard_dim_id, num_shards)
            comb = try_combination(sharded, global_out)
        except HaloHint:
            halo_hinted = True
        except Exception:
            return None
        if comb is not None:
            return comb
        if not allow_halo or not (halo_hinted or self.try_halo):
            return None
        # halo only makes sense when it is smaller than a shard: otherwise
        # every "shard" degenerates to (nearly) the whole tensor and the
        # identity check passes vacuously
        min_base = min(
            (self.flat_tensors[i].shape[d] // num_shards
             for i, dims in enumerate(annotation.annotation)
             for d, sd in enumerate(dims) if sd.shard_dim_id == shard_dim_id),
            default=0)
        # retry with halo-widened input shards
        for width in range(1, mdconfig.max_halo + 1):
            if width >= min_base:
                break
            ann_h = annotation.inject_haloinfo(width, shard_dim_id)
            try:
                sharded = self.exec_sharded(ann_h, shard_dim_id, num_shards)
                comb = try_combination(sharded, global_out)
            except Exception:
                comb = None
            if comb is not None:
                for i in range(len(annotation)):
                    annotation[i] = ann_h[i]
                return comb
        return None

    def sharding_discovery(self):
        """Search shard groups; returns (ShardAnnotation, {id: combination}).

        Every tensor-dim either joins a group (gets a positive shard_dim_id)
        or stays NoShardDim. Dims whose size < num_shards are skipped.
        """
        num_shards = mdconfig.discovery_num_shards
        shapes = [t.shape for t in self.flat_tensors]
        ann = ShardAnnotation.init_from_input_shapes(shapes)
        combination_ann: Dict[int, object] = {}
        try:
            global_out = self.exec_global()
        except Exception as e:
            logger.debug("discovery: global exec failed for %s: %s", self.name, e)
            return ann, combination_ann

        next_id = 1
        all_dims = [(i, d) for i, s in enumerate(shapes) for d in range(len(s))]

        def shardable(i, d):
            return (shapes[i][d] >= num_shards
                    and ann[i][d].shard_dim_id == 0)

        for (i, d) in all_dims:
            if not shardable(i, d):
                continue
            found_comb = None
            group_ann = None
            # 1) seed alone
            trial = ann.copy()
            trial[i][d] = ShardDim.get_shard_dim(next_id)
            comb = self._group_works(trial, next_id, num_shards, global_out)
            if comb is not None:
                found_comb, group_ann = comb, trial
            else:
                # 2) seed + one partner dim from ONE other input
                for (j, e) in all_dims:
                    if j == i or not shardable(j, e):
                        continue
                    trial2 = ann.copy()
                    trial2[i][d] = ShardDim.get_shard_dim(next_id)
                    trial2[j][e] = ShardDim.get_shard_dim(next_id)
                    comb = self._group_works(trial2, next_id, num_shards, global_out)
                    if comb is not None:
                        found_comb, group_ann = comb, trial2
                        break
                if found_comb is None:
                    # 3-way+ groups (e.g. batched attention: q,k,v batch dims
                    # must shard together): tag the SAME dim in every input
                    # that has it with matching size
                    trial3 = ann.copy()
                    trial3[i][d] = ShardDim.get_shard_dim(next_id)
                    tags = 1
                    for j in range(len(shapes)):
                        if j == i or d >= len(shapes[j]):
                            continue
                        if shapes[j][d] == shapes[i][d] and shardable(j, d):
                            trial3[j][d] = ShardDim.get_shard_dim(next_id)
                            tags += 1
                    if tags >= 3:
                        comb = self._group_works(trial3, next_id, num_shards,
                                                 global_out)
                        if comb is not None:
                            found_comb, group_ann = comb, trial3
            if found_comb is None:
                continue
            # 3) greedy extension with remaining dims (one per input)
            for (j, e) in all_dims:
                if not (shapes[j][e] >= num_shards
                        and group_ann[j][e].shard_dim_id == 0):
                    continue
                if any(sd.shard_dim_id == next_id for sd in group_ann[j]):
                    continue  # input already contributes a dim to this group
                trial3 = group_ann.copy()
                trial3[j][e] = ShardDim.get_shard_dim(
                    next_id, halo=max(sd.halo for dims in group_ann.annotation
                                      for sd in dims if sd.shard_dim_id == next_id))
                comb = self._group_works(trial3, next_id, num_shards, global_out,
                                         allow_halo=False)
                if comb is not None and _comb_equal(comb, found_comb):
                    group_ann = trial3
            ann = group_ann
            combination_ann[next_id] = found_comb
            next_id += 1
        return ann, combination_ann


def _comb_equal(a, b) -> bool:
    """Structural equality of combination results (partials or lists)."""
    import functools
    if a is None or b is None:
        return a is b
    if isinstance(a, list) and isinstance(b, list):
        return len(a) == len(b) and all(_comb_equal(x, y) for x, y in zip(a, b))
    if isinstance(a, functools.partial) and isinstance(b, functools.partial):
        return a.func is b.func and a.keywords == b.keywords
    return a == b
