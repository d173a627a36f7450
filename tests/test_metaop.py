"""MetaOp.sharding_discovery on single aten ops (CPU only).

Mirrors the reference's tests/test_unfiyshard/test_unifyop.py coverage.
"""
import torch

from easydist_amd.metashard.combination import CombinationFunc
from easydist_amd.metashard.metaop import MetaOp


def _ids(ann):
    return [[sd.shard_dim_id for sd in dims] for dims in ann.annotation]


def test_matmul_discovery():
    op = MetaOp(torch.mm, (torch.randn(8, 6), torch.randn(6, 4)), name="mm")
    ann, combs = op.sharding_discovery()
    ids = _ids(ann)
    # three groups: A-rows (gather 0), contraction (reduce), B-cols (gather 1)
    assert ids[0][0] != 0 and ids[0][1] != 0 and ids[1][1] != 0
    assert ids[0][1] == ids[1][0]  # contraction dims grouped
    kinds = {}
    for sid, c in combs.items():
        kinds[sid] = (c.func.__name__, c.keywords.get("dim"))
    assert kinds[ids[0][0]] == ("gather", 0)
    assert kinds[ids[1][1]] == ("gather", 1)
    assert combs[ids[0][1]].func is CombinationFunc.reduce


def test_add_broadcast():
    op = MetaOp(torch.add, (torch.randn(8, 6), torch.randn(6)), name="add")
    ann, combs = op.sharding_discovery()
    ids = _ids(ann)
    assert ids[0][0] != 0          # batch dim alone
    assert ids[0][1] == ids[1][0]  # feature dim grouped with bias


def test_relu():
    op = MetaOp(torch.relu, (torch.randn(8, 6),), name="relu")
    ann, combs = op.sharding_discovery()
    ids = _ids(ann)
    assert ids[0][0] != 0 and ids[0][1] != 0
    for c in combs.values():
        assert c.func is CombinationFunc.gather


def test_sum_dim():
    op = MetaOp(lambda x: torch.sum(x, dim=1), (torch.randn(8, 6),), name="sum")
    ann, combs = op.sharding_discovery()
    ids = _ids(ann)
    assert ids[0][0] != 0
    assert combs[ids[0][0]].keywords.get("dim") == 0
    # summing dim itself -> partial
    assert ids[0][1] != 0
    assert combs[ids[0][1]].func is CombinationFunc.reduce


def test_layer_norm():
    w, b = torch.randn(6), torch.randn(6)
    op = MetaOp(lambda x, w, b: torch.nn.functional.layer_norm(x, (6,), w, b),
                (torch.randn(8, 6), w, b), name="ln")
    ann, combs = op.sharding_discovery()
    ids = _ids(ann)
    assert ids[0][0] != 0  # batch shardable
    assert ids[0][1] == 0  # normalized dim not shardable


def test_conv_halo():
    x = torch.randn(2, 3, 16, 16)
    w = torch.randn(4, 3, 3, 3)
    op = MetaOp(lambda x, w: torch.nn.functional.conv2d(x, w, padding=1),
                (x, w), name="aten.convolution")
    ann, combs = op.sharding_discovery()
    ids = _ids(ann)
    assert ids[0][0] != 0  # batch
    # spatial dim shardable with halo=1
    sid = ids[0][2]
    assert sid != 0
    assert ann[0][2].halo == 1


def test_softmax():
    op = MetaOp(lambda x: torch.softmax(x, dim=-1), (torch.randn(8, 6),),
                name="softmax")
    ann, combs = op.sharding_discovery()
    ids = _ids(ann)
    assert ids[0][0] != 0
    assert ids[0][1] == 0  # softmax dim not shardable
