"""Rectangle-intersection P2P reshard (reference
sharding.py:336-612): S(i)/S(j) layouts across a 2x2 mesh move in ONE
batched pairwise exchange instead of per-dim collective chains."""
import pytest
import torch

from easydist_amd.utils.testing import spawn


def _body(world_size):
    import torch.distributed as dist

    from easydist_amd import easydist_setup, set_device_mesh
    from easydist_amd.parallel import comm
    from easydist_amd.parallel.device_mesh import get_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([[0, 1], [2, 3]], ["spmd0", "spmd1"])
    mesh = get_device_mesh()
    r = dist.get_rank()
    torch.manual_seed(0)
    g = torch.randn(8, 12)
    dist.broadcast(g, src=0)

    def local_of(t, placements):
        coords = [r // 2, r % 2]
        rect = comm._rect_of(list(t.shape), placements, coords, (2, 2))
        return t[tuple(slice(lo, hi) for lo, hi in rect)].contiguous()

    cases = [
        ([("S", 0), ("S", 1)], [("S", 1), ("S", 0)]),   # transpose layout
        ([("S", 0), ("R",)], [("S", 1), ("S", 0)]),
        ([("S", 0), ("S", 0)], [("R",), ("S", 1)]),
        ([("S", 1), ("S", 1)], [("S", 0), ("S", 0)]),
    ]
    for cur, want in cases:
        t = local_of(g, cur)
        out = comm.p2p_reshard(t, list(g.shape), cur, want, mesh)
        ref = local_of(g, want)
        assert torch.equal(out, ref), (cur, want, r)


@pytest.mark.world4
def test_p2p_reshard_2x2():
    spawn(_body, args=(4,), world_size=4, port=29568)
