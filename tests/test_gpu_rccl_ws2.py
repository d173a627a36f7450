"""RCCL world-2 collective paths over real RCCL.

VERDICT item 6 suggested two ranks sharing ONE MI355X; RCCL 2.26 rejects
that outright ("Duplicate GPU detected: rank 0 and rank 1 both on CUDA
device" — ncclInvalidUsage), so these tests require >= 2 visible GPUs
and run whenever the driver lands on a multi-GPU box; the gloo goldens
cover the same graph-level paths every round on CPU."""
import copy

import pytest
import torch

from easydist_amd.utils.testing import spawn

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available() or torch.cuda.device_count() < 2,
    reason="RCCL refuses 2 ranks on one device (Duplicate GPU detected); "
           "needs >= 2 MI355X")


def _collectives_body(world_size):
    import torch.distributed as dist

    from easydist_amd.parallel import comm

    r = dist.get_rank()
    dev = torch.device("cuda")
    t = torch.full((4, 8), float(r + 1), device=dev)
    ag = comm.all_gather(t, 0, None)
    assert ag.shape == (8, 8)
    assert float(ag[0, 0]) == 1.0 and float(ag[4, 0]) == 2.0

    rs = comm.reduce_scatter(torch.ones(8, 4, device=dev) * (r + 1), 0,
                             "sum", None)
    assert rs.shape == (4, 4)
    assert float(rs[0, 0]) == 3.0

    x = torch.arange(8.0, device=dev).reshape(8, 1) + r * 100
    a2a = comm.all_to_all(x, 0, 1, None)
    assert a2a.shape == (4, 2)

    ar = torch.ones(4, device=dev) * (r + 1)
    out = comm.all_reduce(ar, "sum", None)
    assert float(out[0]) == 3.0


@requires_gpu
def test_rccl_collectives_ws2():
    spawn(_collectives_body, args=(2,), world_size=2, port=29591,
          backend="nccl")


def _golden_body(world_size):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
    from easydist_amd.models import gpt as gptm
    from dataclasses import replace

    easydist_setup(backend="torch", device="cuda")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = replace(gptm.GPT2_SMALL, n_layer=2, n_embd=128, n_head=2,
                  block_size=128, vocab_size=512)
    model = gptm.GPT(cfg).cuda()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    opt_ref = torch.optim.Adam(ref.parameters(), lr=1e-3, fused=True)

    def train_step(model, opt, idx, targets):
        return gptm.gpt_train_step(model, opt, idx, targets)

    compiled = easydist_compile(train_step, parallel_mode="auto")
    g = torch.Generator().manual_seed(5)
    idx = torch.randint(0, 512, (8, 128), generator=g).cuda()
    tg = torch.randint(0, 512, (8, 128), generator=g).cuda()
    for step in range(3):
        loss = compiled(model, opt, idx, tg)
        rl = train_step(ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(rl)) < 2e-2, (step, float(loss),
                                                     float(rl))


@requires_gpu
def test_rccl_auto_spmd_golden_ws2():
    spawn(_golden_body, args=(2,), world_size=2, port=29592,
          backend="nccl")
