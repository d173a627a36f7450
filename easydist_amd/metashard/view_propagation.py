"""Analytic sharding rules for view/reshape ops.

Capability parity with reference ``easydist/metashard/view_propagation.py``
(view_propagation / view_propagation_preset, lines 33-129), re-derived.

A view is a reshape ``in_shape -> out_shape`` with numel preserved. We
decompose both shapes into an alignment of "segments": maximal runs where the
products match. Within one segment, the *outermost* factor boundary is where
a contiguous shard survives the reshape: sharding input dim i maps to
sharding output dim o iff the elements of the shard stay contiguous, which
holds when the prefix product up to i in the input equals the prefix product
up to o in the output and in_shape[i] % n == 0 keeps block alignment
(divisibility is checked later by the strategy pool).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple


def _squeeze_ones(shape):
    """Return (squeezed shape, map squeezed-idx -> original idx)."""
    out, idx = [], []
    for i, s in enumerate(shape):
        if s != 1:
            out.append(s)
            idx.append(i)
    return out, idx


def view_dim_map(in_shape: List[int], out_shape: List[int]) -> Dict[int, Tuple[int, int]]:
    """Map shardable input dims to output dims for a reshape.

    Returns {in_dim: (out_dim, inner_block)} where ``inner_block`` is the
    number of elements that one index step of in_dim covers inside out_dim
    (1 when the dims align exactly at their leading edge). Sharding in_dim
    into n contiguous pieces equals sharding out_dim into n contiguous pieces
    iff in_dim is the OUTERMOST non-trivial factor of its segment — which is
    exactly when its prefix products align; only those dims appear here.
    """
    in_sq, in_idx = _squeeze_ones(in_shape)
    out_sq, out_idx = _squeeze_ones(out_shape)

    mapping: Dict[int, Tuple[int, int]] = {}
    i = j = 0
    # walk segments of equal product
    while i < len(in_sq) and j < len(out_sq):
        seg_in_start, seg_out_start = i, j
        pi, pj = in_sq[i], out_sq[j]
        i += 1
        j += 1
        while pi != pj:
            if pi < pj:
                pi *= in_sq[i]
                i += 1
            else:
                pj *= out_sq[j]
                j += 1
        # segment [seg_in_start, i) <-> [seg_out_start, j)
        # the leading input dim of the segment maps to the leading output dim
        lead_in = in_idx[seg_in_start]
        lead_out = out_idx[seg_out_start]
        # inner block size = elements per index step of lead_in inside lead_out
        in_block = 1
        for k in range(seg_in_start + 1, i):
            in_block *= in_sq[k]
        out_block = 1
        for k in range(seg_out_start + 1, j):
            out_block *= out_sq[k]
        mapping[lead_in] = (lead_out, in_block // out_block if out_block and
                            in_block % out_block == 0 else 0)
        # a 1:1 tail inside the segment also maps (e.g. [a,b]->[a,b])
        if i - seg_in_start == j - seg_out_start:
            ok = all(in_sq[seg_in_start + k] == out_sq[seg_out_start + k]
                     for k in range(i - seg_in_start))
            if ok:
                for k in range(1, i - seg_in_start):
                    mapping[in_idx[seg_in_start + k]] = (out_idx[seg_out_start + k], 1)
    return mapping


def local_view_shape(out_shape: List[int], out_dim: int, num_shards: int):
    """The local (per-rank) shape argument for a view whose out_dim is sharded."""
    new_shape = list(out_shape)
    if new_shape[out_dim] == -1:
        return new_shape
    assert new_shape[out_dim] % num_shards == 0, (
        f"view dim {out_dim} size {new_shape[out_dim]} not divisible by {num_shards}")
    new_shape[out_dim] = new_shape[out_dim] // num_shards
    return new_shape


def view_propagation(in_shape: List[int], out_shape: List[int]):
    """Produce (sharding_ann-like, combination-like) description for a view.

    Returns dict: {'in_dim_to_out_dim': {i: o}} restricted to dims that
    shard cleanly (leading-edge alignment).
    """
    # resolve a single -1 in out_shape
    out_shape = list(out_shape)
    numel = 1
    for s in in_shape:
        numel *= s
    if -1 in out_shape:
        known = 1
        for s in out_shape:
            if s != -1:
                known *= s
        out_shape[out_shape.index(-1)] = numel // known if known else 0
    m = view_dim_map(list(in_shape), out_shape)
    return {i: o for i, (o, blk) in m.items()}
