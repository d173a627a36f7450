"""End-to-end auto-SPMD golden tests (CPU, gloo).

The reference's golden pattern (tests/test_torch/test_spmd.py:60-117): build
the model twice, run vanilla torch vs the compiled version for several
steps, assert params/opt-states/loss match.
"""
import copy

import pytest
import torch
import torch.nn as nn

from easydist_amd.utils.testing import init_single_process, spawn


class MLP(nn.Module):
    def __init__(self, d=16, h=32):
        super().__init__()
        self.fc1 = nn.Linear(d, h)
        self.norm = nn.LayerNorm(h)
        self.fc2 = nn.Linear(h, d)

    def forward(self, x):
        return self.fc2(self.norm(torch.relu(self.fc1(x))))


def train_step(model, opt, x, y):
    loss = ((model(x) - y) ** 2).mean()
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss


def _run_golden(world_size):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])

    torch.manual_seed(42)
    model = MLP()
    # broadcast initial weights so every rank starts identical
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)

    opt = torch.optim.Adam(model.parameters(), lr=1e-2, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-2, fused=True)

    compiled = easydist_compile(train_step, parallel_mode="auto",
                                cuda_graph=False)

    torch.manual_seed(7)
    losses, ref_losses = [], []
    for step in range(4):
        x = torch.randn(8, 16)
        y = torch.randn(8, 16)
        dist.broadcast(x, src=0)
        dist.broadcast(y, src=0)
        loss = compiled(model, opt, x, y)
        ref_loss = train_step(model_ref, opt_ref, x, y)
        losses.append(float(loss))
        ref_losses.append(float(ref_loss))
    for l, rl in zip(losses, ref_losses):
        assert abs(l - rl) < 1e-4, (losses, ref_losses)
    # params must match after training
    final = compiled.named_parameters()
    for n, p_ref in model_ref.named_parameters():
        got = final[n]
        assert torch.allclose(got, p_ref.detach(), rtol=1e-4, atol=1e-5), \
            (n, (got - p_ref.detach()).abs().max())


def test_auto_spmd_ws1():
    init_single_process()
    _run_golden(1)


@pytest.mark.world2
def test_auto_spmd_ws2():
    spawn(_run_golden, args=(2,), world_size=2, port=29532)


def test_auto_state_dict_roundtrip():
    """Checkpoint-and-replay: losses after reload must match exactly."""
    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh

    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])
    torch.manual_seed(0)
    model = MLP()
    opt = torch.optim.Adam(model.parameters(), lr=1e-2, fused=True)
    compiled = easydist_compile(train_step, cuda_graph=False)
    torch.manual_seed(3)
    batches = [(torch.randn(8, 16), torch.randn(8, 16)) for _ in range(4)]
    for x, y in batches[:2]:
        compiled(model, opt, x, y)
    rt = list(compiled.compiled.values())[0]
    ckpt = rt.state_dict()
    assert any(k == "fc1.weight" for k in ckpt), list(ckpt)[:4]
    assert any(k.endswith(".exp_avg") for k in ckpt), list(ckpt)[:8]
    later = [float(compiled(model, opt, x, y)) for x, y in batches[2:]]
    rt.load_state_dict(ckpt)
    replay = [float(compiled(model, opt, x, y)) for x, y in batches[2:]]
    assert later == replay, (later, replay)


def _run_golden_2d(world_size):
    """2-D SPMD mesh (2x2): the per-mesh-dim solver composes placements
    across both dims; golden vs vanilla."""
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([[0, 1], [2, 3]], ["spmd0", "spmd1"])

    torch.manual_seed(42)
    model = MLP()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-2, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-2, fused=True)
    compiled = easydist_compile(train_step, parallel_mode="auto",
                                cuda_graph=False)
    torch.manual_seed(7)
    for step in range(3):
        x = torch.randn(8, 16)
        y = torch.randn(8, 16)
        dist.broadcast(x, src=0)
        dist.broadcast(y, src=0)
        loss = compiled(model, opt, x, y)
        ref_loss = train_step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref_loss)) < 1e-4, \
            (step, float(loss), float(ref_loss))
    final = compiled.named_parameters()
    for n, p_ref in model_ref.named_parameters():
        got = final[n]
        assert torch.allclose(got, p_ref.detach(), rtol=1e-4, atol=1e-5), \
            (n, (got - p_ref.detach()).abs().max())


@pytest.mark.world4
def test_auto_spmd_2d_mesh_ws4():
    spawn(_run_golden_2d, args=(4,), world_size=4, port=29533)


def _pure_fn_body(world_size):
    """easydist_compile of a plain function — no module, no optimizer
    (reference: tests/test_torch/test_simple.py's fn_2 shape)."""
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])

    def fn(x, y):
        return torch.mm(torch.exp(torch.tanh(x)), y)

    c = easydist_compile(fn, cuda_graph=False)
    torch.manual_seed(0)
    x = torch.randn(8, 8)
    y = torch.randn(8, 8)
    dist.broadcast(x, src=0)
    dist.broadcast(y, src=0)
    out = c(x, y)
    assert torch.allclose(out, fn(x, y), rtol=1e-5, atol=1e-6)


def test_pure_function_ws1():
    init_single_process()
    _pure_fn_body(1)


@pytest.mark.world2
def test_pure_function_ws2():
    spawn(_pure_fn_body, args=(2,), world_size=2, port=29534)


def _beam_body(world_size):
    """Beam-search solver path (used automatically above
    ilp_max_clusters): golden vs vanilla on gloo ws2."""
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, mdconfig, \
        set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    mdconfig.solver_mode = "beam"
    mdconfig.enable_compile_cache = False   # force a real beam solve
    try:
        torch.manual_seed(42)
        model = MLP()
        for p in model.parameters():
            dist.broadcast(p.data, src=0)
        model_ref = copy.deepcopy(model)
        opt = torch.optim.Adam(model.parameters(), lr=1e-2, fused=True)
        opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-2,
                                   fused=True)
        compiled = easydist_compile(train_step, cuda_graph=False)
        torch.manual_seed(7)
        for step in range(3):
            x = torch.randn(8, 16)
            y = torch.randn(8, 16)
            dist.broadcast(x, src=0)
            dist.broadcast(y, src=0)
            loss = compiled(model, opt, x, y)
            ref = train_step(model_ref, opt_ref, x, y)
            assert abs(float(loss) - float(ref)) < 1e-4, \
                (step, float(loss), float(ref))
    finally:
        mdconfig.solver_mode = "ilp"
        mdconfig.enable_compile_cache = True


@pytest.mark.world2
def test_beam_solver_ws2():
    spawn(_beam_body, args=(2,), world_size=2, port=29535)
