"""MoE expert-parallel tests (CPU, gloo): the EP all-to-all dispatch must
reproduce the single-process MoE exactly (same weights, same tokens)."""
import copy

import pytest
import torch

from easydist_amd.models.moe import MIXTRAL_SMALL, MoEConfig, MoELayer
from easydist_amd.utils.testing import init_single_process, spawn


def test_moe_layer_local_forward():
    torch.manual_seed(0)
    cfg = MIXTRAL_SMALL
    layer = MoELayer(cfg)
    x = torch.randn(2, 16, cfg.n_embd)
    out = layer(x)
    assert out.shape == x.shape
    assert torch.isfinite(out).all()
    # gradient flows to every expert weight and the router
    out.sum().backward()
    assert layer.router.weight.grad is not None
    assert layer.experts.w1.grad is not None


def test_moe_capacity_determinism():
    torch.manual_seed(0)
    cfg = MoEConfig(vocab_size=64, n_layer=1, n_head=2, n_embd=32,
                    block_size=8, n_experts=4, top_k=2, ffn_hidden=64)
    layer = MoELayer(cfg)
    x = torch.randn(1, 8, 32)
    a = layer(x)
    b = layer(x)
    assert torch.equal(a, b)


def _ep_body(world_size):
    import torch.distributed as dist

    torch.manual_seed(0)
    cfg = MoEConfig(vocab_size=64, n_layer=1, n_head=2, n_embd=32,
                    block_size=8, n_experts=4, top_k=2, ffn_hidden=64,
                    capacity_factor=8.0)   # no drops: exact comparison
    ref = MoELayer(cfg)          # all experts in one place
    for p in ref.parameters():
        dist.broadcast(p.data, src=0)

    ep = MoELayer(cfg, ep_group=dist.group.WORLD)
    # load my shard of the expert bank + full router
    L = cfg.n_experts // world_size
    r = dist.get_rank()
    with torch.no_grad():
        ep.router.weight.copy_(ref.router.weight)
        ep.experts.w1.copy_(ref.experts.w1[r * L:(r + 1) * L])
        ep.experts.w3.copy_(ref.experts.w3[r * L:(r + 1) * L])
        ep.experts.w2.copy_(ref.experts.w2[r * L:(r + 1) * L])

    torch.manual_seed(11)
    x = torch.randn(2, 8, 32)
    dist.broadcast(x, src=0)
    # each rank processes its batch shard; ref processes the full batch
    shard = torch.chunk(x, world_size, dim=0)[r]
    out_ep = ep(shard)
    out_ref = torch.chunk(ref(x), world_size, dim=0)[r]
    assert torch.allclose(out_ep, out_ref, rtol=1e-4, atol=1e-5), \
        (out_ep - out_ref).abs().max()


@pytest.mark.world2
def test_moe_ep_ws2():
    spawn(_ep_body, args=(2,), world_size=2, port=29561)


def _ep_grad_body(world_size):
    """Grads must flow through dispatch AND combine all-to-alls: the EP
    expert-weight grad shard (after all-reducing over the DP dimension of
    the data split) must equal the single-process grad."""
    import torch.distributed as dist

    torch.manual_seed(0)
    cfg = MoEConfig(vocab_size=64, n_layer=1, n_head=2, n_embd=32,
                    block_size=8, n_experts=4, top_k=2, ffn_hidden=64,
                    capacity_factor=8.0)
    ref = MoELayer(cfg)
    for p in ref.parameters():
        dist.broadcast(p.data, src=0)
    ep = MoELayer(cfg, ep_group=dist.group.WORLD)
    L = cfg.n_experts // world_size
    r = dist.get_rank()
    with torch.no_grad():
        ep.router.weight.copy_(ref.router.weight)
        ep.experts.w1.copy_(ref.experts.w1[r * L:(r + 1) * L])
        ep.experts.w3.copy_(ref.experts.w3[r * L:(r + 1) * L])
        ep.experts.w2.copy_(ref.experts.w2[r * L:(r + 1) * L])

    torch.manual_seed(11)
    x = torch.randn(2, 8, 32)
    dist.broadcast(x, src=0)
    shard = torch.chunk(x, world_size, dim=0)[r]
    # sum-loss so per-token grads just add across the batch split
    ep(shard).sum().backward()
    ref(x).sum().backward()
    # expert grads are COMPLETE locally: each expert saw every rank's
    # tokens through the dispatch all-to-all
    for wname in ("w1", "w3", "w2"):
        g = getattr(ep.experts, wname).grad
        want = getattr(ref.experts, wname).grad[r * L:(r + 1) * L]
        assert torch.allclose(g, want, rtol=1e-4, atol=1e-5), \
            (wname, (g - want).abs().max())
    # router grads cover only local tokens: sum over the batch split
    gr = ep.router.weight.grad.clone()
    dist.all_reduce(gr)
    assert torch.allclose(gr, ref.router.weight.grad, rtol=1e-4,
                          atol=1e-5)


@pytest.mark.world2
def test_moe_ep_grads_ws2():
    spawn(_ep_grad_body, args=(2,), world_size=2, port=29562)


def _moe_auto_body(world_size):
    """Full auto-SPMD compile of the MoE model, golden vs vanilla."""
    import copy

    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.moe import MIXTRAL_SMALL, MoEGPT

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = MIXTRAL_SMALL
    model = MoEGPT(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3,
                               fused=True)

    def step(model, opt, idx, tg):
        loss = model.loss(idx, tg)
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    compiled = easydist_compile(step, cuda_graph=False)
    torch.manual_seed(5)
    for i in range(2):
        idx = torch.randint(0, cfg.vocab_size, (4, cfg.block_size))
        tg = torch.randint(0, cfg.vocab_size, (4, cfg.block_size))
        dist.broadcast(idx, src=0)
        dist.broadcast(tg, src=0)
        loss = compiled(model, opt, idx, tg)
        ref = step(model_ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref)) < 5e-3, \
            (i, float(loss), float(ref))


def test_moe_auto_ws1():
    from easydist_amd.utils.testing import init_single_process
    init_single_process()
    _moe_auto_body(1)


@pytest.mark.world2
def test_moe_auto_ws2():
    spawn(_moe_auto_body, args=(2,), world_size=2, port=29565)


@pytest.mark.world4
def test_moe_auto_ws4():
    spawn(_moe_auto_body, args=(4,), world_size=4, port=29662)
