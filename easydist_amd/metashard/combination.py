"""Combination functions + the search that identifies them.

Capability parity with the reference's ``easydist/metashard/combination.py``
(CombinationFunc identity/reduce/gather incl. halo and chunk variants,
try_combination; reference lines 140-313). Torch-only re-design.

``try_combination(sharded_out, global_out)`` answers: given the op's outputs
when run per-shard, which function of the shard outputs reproduces the global
output? The answer (identity / reduce(sum|max|min) / gather(dim)) maps
directly onto SPMD placements: identity -> REPLICATE, reduce -> PARTIAL(op),
gather(dim) -> SHARD(dim).
"""
from __future__ import annotations

import functools
import operator
from typing import List, Optional

import torch

from .. import config as mdconfig


def _allclose(a: torch.Tensor, b: torch.Tensor) -> bool:
    if a.shape != b.shape:
        return False
    if a.dtype != b.dtype:
        return False
    if not a.is_floating_point():
        return bool(torch.equal(a, b))
    if a.dtype == torch.float64:
        rtol, atol = mdconfig.discovery_rtol, mdconfig.discovery_atol
    else:
        rtol = mdconfig.discovery_rtol_lowprec
        atol = mdconfig.discovery_atol_lowprec
    return bool(torch.allclose(a.double(), b.double(), rtol=rtol, atol=atol,
                               equal_nan=True))


class CombinationFunc:
    """The recombination primitives. Each is usable as functools.partial."""

    @staticmethod
    def identity(shards: List[torch.Tensor]) -> torch.Tensor:
        return shards[0]

    @staticmethod
    def reduce(shards: List[torch.Tensor], ops=operator.add) -> torch.Tensor:
        return functools.reduce(ops, shards)

    @staticmethod
    def reduce_mean(shards: List[torch.Tensor]) -> torch.Tensor:
        """Mean across equal-size shards (RCCL ReduceOp.AVG on the wire)."""
        return functools.reduce(operator.add, shards) / len(shards)

    @staticmethod
    def gather(shards: List[torch.Tensor], dim: int, halowidth: int = 0,
               chunk: int = 1) -> torch.Tensor:
        if halowidth == 0 and chunk == 1:
            return torch.concat(shards, dim=dim)
        if halowidth != 0:
            # trim the overlapping halo region of each interior boundary;
            # each shard carries `halowidth` extra elements on each side that
            # touches a neighbour.
            trimmed = []
            n = len(shards)
            for i, s in enumerate(shards):
                lo = halowidth if i > 0 else 0
                hi = halowidth if i < n - 1 else 0
                idx = [slice(None)] * s.dim()
                idx[dim] = slice(lo, s.shape[dim] - hi if hi else None)
                trimmed.append(s[tuple(idx)])
            return torch.concat(trimmed, dim=dim)
        # block-cyclic: each shard holds `chunk` interleaved blocks
        pieces = [list(torch.chunk(s, chunk, dim=dim)) for s in shards]
        inter = []
        for c in range(chunk):
            for p in pieces:
                inter.append(p[c])
        return torch.concat(inter, dim=dim)


ReduceOp = {
    "sum": operator.add,
    "max": torch.maximum,
    "min": torch.minimum,
}


class HaloHint(Exception):
    """Raised when gather shapes suggest an overlap: retry discovery w/ halo."""

    def __init__(self, dim):
        super().__init__(f"halo hint on dim {dim}")
        self.dim = dim


def _try_combination_single(shards: List[torch.Tensor],
                            global_out: torch.Tensor,
                            allow_halo_hint: bool = True):
    """Find the combination for one output tensor. Returns a partial or None."""
    if not isinstance(global_out, torch.Tensor):
        # non-tensor output: all shards must equal the global value
        if all(s == global_out for s in shards):
            return functools.partial(CombinationFunc.identity)
        return None

    same_shape = all(s.shape == global_out.shape for s in shards)
    if same_shape:
        if all(_allclose(s, global_out) for s in shards):
            return functools.partial(CombinationFunc.identity)
        if global_out.is_floating_point() or global_out.dtype in (
                torch.int32, torch.int64):
            for name, op in ReduceOp.items():
                try:
                    if _allclose(CombinationFunc.reduce(shards, ops=op), global_out):
                        return functools.partial(CombinationFunc.reduce, ops=op)
                except Exception:
                    continue
            try:
                if (len({tuple(s.shape) for s in shards}) == 1
                        and _allclose(CombinationFunc.reduce_mean(shards),
                                      global_out)):
                    return functools.partial(CombinationFunc.reduce_mean)
            except Exception:
                pass

    # gather candidates: dims where per-shard sizes sum to the global size
    if all(s.dim() == global_out.dim() for s in shards):
        for dim in range(global_out.dim()):
            other_ok = all(
                all(s.shape[d] == global_out.shape[d]
                    for d in range(global_out.dim()) if d != dim)
                for s in shards)
            if not other_ok:
                continue
            total = sum(s.shape[dim] for s in shards)
            if total == global_out.shape[dim]:
                try:
                    if _allclose(CombinationFunc.gather(shards, dim=dim), global_out):
                        return functools.partial(CombinationFunc.gather, dim=dim)
                except Exception:
                    pass
            elif total > global_out.shape[dim] and allow_halo_hint:
                extra = total - global_out.shape[dim]
                n = len(shards)
                if n > 1 and extra % (2 * (n - 1)) == 0:
                    width = extra // (2 * (n - 1))
                    if 0 < width <= mdconfig.max_halo:
                        # shards overlap: try halo-trimmed gather right away
                        try:
                            cand = CombinationFunc.gather(shards, dim=dim,
                                                          halowidth=width)
                            if _allclose(cand, global_out):
                                return functools.partial(CombinationFunc.gather,
                                                         dim=dim, halowidth=width)
                        except Exception:
                            pass
                        raise HaloHint(dim)
    return None


def try_combination(sharded_out, global_out):
    """Find a combination rule for (possibly nested) op outputs.

    Args:
        sharded_out: list over shards, each the op's (pytree) output.
        global_out: the unsharded op output.

    Returns:
        a matching pytree of partial(CombinationFunc...) (or a single partial
        for single-tensor outputs), or None when no rule reproduces the
        global output.
    """
    if isinstance(global_out, torch.Tensor):
        return _try_combination_single(list(sharded_out), global_out)
    if isinstance(global_out, (tuple, list)):
        combs = []
        for i, g in enumerate(global_out):
            if g is None:
                if all(s[i] is None for s in sharded_out):
                    combs.append(None)
                    continue
                return None
            comb = try_combination([s[i] for s in sharded_out], g)
            if comb is None and isinstance(g, torch.Tensor):
                return None
            combs.append(comb)
        return combs
    # scalar output
    return _try_combination_single(list(sharded_out), global_out)
