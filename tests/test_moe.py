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


# ---------------------------------------------------------------------------
# EP THROUGH THE COMPILER: the auto-SPMD solver chooses
# the expert-parallel strategy for the opaque routing ops and the
# sharding transform emits the S(cap)->S(expert) all-to-alls itself.
# ---------------------------------------------------------------------------
def _auto_ep_body(world_size):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
    from easydist_amd.models import moe as moem

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    r = dist.get_rank()

    torch.manual_seed(0)
    cfg = moem.MoEConfig(vocab_size=64, n_layer=1, n_head=2, n_embd=64,
                         block_size=32, n_experts=4, top_k=2, ffn_hidden=256,
                         capacity_factor=8.0)   # no drops: exact comparison
    model = moem.MoEGPT(cfg, ep_group=None)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, foreach=False)
    opt_ref = torch.optim.Adam(ref.parameters(), lr=1e-3, foreach=False)

    def train_step(model, opt, idx, targets):
        return moem.moe_train_step(model, opt, idx, targets)

    compiled = easydist_compile(train_step, parallel_mode="auto")

    g = torch.Generator().manual_seed(7)
    B = 2 * world_size
    idx = torch.randint(0, cfg.vocab_size, (B, cfg.block_size), generator=g)
    tg = torch.randint(0, cfg.vocab_size, (B, cfg.block_size), generator=g)

    for step in range(2):
        loss = compiled(model, opt, idx, tg)
        ref_loss = train_step(ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref_loss)) < 5e-3, \
            (step, float(loss), float(ref_loss))

    gm = list(compiled.compiled.values())[0].gm
    names = [getattr(n.target, "__name__", "") for n in gm.graph.nodes
             if n.op == "call_function"]
    # the routing runs through the compiler-visible ops; at this (small)
    # scale the solver may legitimately prefer replicated routing — the
    # EP all-to-all choice itself is asserted at realistic scale in
    # test_moe_auto_ep_solver_ws2
    assert names.count("moe_bins.default") == 1
    assert names.count("moe_combine.default") == 1


@pytest.mark.world2
def test_moe_auto_ep_ws2():
    spawn(_auto_ep_body, args=(2,), world_size=2, port=29563)


def _auto_ep_solver_body(world_size):
    """Solve-only (no execution): at Mixtral-like scale the solver must
    choose expert parallelism — rt_all_to_all reshards around the expert
    FFN — because replicating the expert bank (or all-reducing its
    gradients) is costlier than exchanging the token bins over xGMI."""
    import torch.distributed as dist

    from easydist_amd import easydist_setup, set_device_mesh
    from easydist_amd.compiler.compile_auto import shard_graph
    from easydist_amd.compiler.passes.functionalize import canonicalize
    from easydist_amd.compiler.tracing import ed_compile_func
    from easydist_amd.models import moe as moem
    from easydist_amd.parallel.device_mesh import get_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = moem.MoEConfig(vocab_size=4096, n_layer=1, n_head=8, n_embd=1024,
                         block_size=512, n_experts=8, top_k=2,
                         ffn_hidden=4096, capacity_factor=1.25)
    model = moem.MoEGPT(cfg, ep_group=None)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, foreach=False)

    def train_step(model, opt, idx, targets):
        return moem.moe_train_step(model, opt, idx, targets)

    idx = torch.randint(0, cfg.vocab_size, (8, 512))
    tg = torch.randint(0, cfg.vocab_size, (8, 512))
    params, buffers, named_states, gm = ed_compile_func(
        train_step, "fake", (model, opt, idx, tg), {}, model, opt)
    gm, io_map = canonicalize(gm)
    gm2, env, _, _ = shard_graph(gm, get_device_mesh(), io_map, set(), "cpu")
    names = [getattr(n.target, "__name__", "") for n in gm2.graph.nodes
             if n.op == "call_function"]
    n_a2a = sum(1 for s in names if s == "rt_all_to_all_start")
    assert n_a2a >= 2, (n_a2a, [s for s in names if s.startswith("rt_")])


@pytest.mark.world2
def test_moe_auto_ep_solver_ws2():
    spawn(_auto_ep_solver_body, args=(2,), world_size=2, port=29564)
