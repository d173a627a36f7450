"""Unit tests for the ShardCombine combination algebra (CPU only).

Mirrors the coverage shape of the reference's tests/test_combination
(identity / reduce / gather / halo / chunk variants).
"""
import functools
import operator

import pytest
import torch

from easydist_amd.metashard.combination import (CombinationFunc, HaloHint,
                                                try_combination)
from easydist_amd.metashard.metaop import shard_tensor


def test_gather_simple():
    t = torch.randn(8, 6)
    shards = shard_tensor(t, 0, 2)
    comb = try_combination(shards, t)
    assert comb is not None and comb.func is CombinationFunc.gather
    assert comb.keywords["dim"] == 0
    assert torch.equal(comb(shards), t)


def test_gather_dim1():
    t = torch.randn(4, 12)
    shards = shard_tensor(t, 1, 3)
    comb = try_combination(shards, t)
    assert comb.keywords["dim"] == 1


def test_identity():
    t = torch.randn(5, 5)
    comb = try_combination([t.clone(), t.clone()], t)
    assert comb.func is CombinationFunc.identity


def test_reduce_sum():
    a, b = torch.randn(4, 4), torch.randn(4, 4)
    comb = try_combination([a, b], a + b)
    assert comb.func is CombinationFunc.reduce
    assert comb.keywords["ops"] is operator.add


def test_reduce_max():
    a, b = torch.randn(4, 4), torch.randn(4, 4)
    comb = try_combination([a, b], torch.maximum(a, b))
    assert comb.func is CombinationFunc.reduce
    assert comb.keywords["ops"] is torch.maximum


def test_halo_gather():
    t = torch.randn(10, 4)
    shards = shard_tensor(t, 0, 2, halo=1)
    assert shards[0].shape[0] == 6 and shards[1].shape[0] == 6
    comb = functools.partial(CombinationFunc.gather, dim=0, halowidth=1)
    assert torch.equal(comb(shards), t)


def test_halo_hint_raised():
    t = torch.randn(10, 4)
    shards = shard_tensor(t, 0, 2, halo=1)
    # overlapping shards whose values DON'T match a clean halo trim
    bad = [s + 1e-1 * torch.randn_like(s) for s in shards]
    with pytest.raises(HaloHint):
        try_combination(bad, t)


def test_chunk_gather():
    t = torch.arange(16).reshape(16, 1).float()
    shards = shard_tensor(t, 0, 2, chunk=2)
    # block-cyclic: shard 0 holds blocks 0 and 2
    assert torch.equal(shards[0][:4], t[:4])
    assert torch.equal(shards[0][4:], t[8:12])
    comb = functools.partial(CombinationFunc.gather, dim=0, chunk=2)
    assert torch.equal(comb(shards), t)


def test_tuple_output():
    a = torch.randn(8, 4)
    shards_in = shard_tensor(a, 0, 2)
    outs = [(s, s.sum(0)) for s in shards_in]
    glob = (a, a.sum(0))
    combs = try_combination(outs, glob)
    assert combs[0].func is CombinationFunc.gather
    assert combs[1].func is CombinationFunc.reduce


def test_no_rule():
    t = torch.randn(8, 4)
    # mismatched shard shapes: no identity/reduce/gather applies
    assert try_combination([torch.randn(3, 4), torch.randn(3, 4)], t) is None
