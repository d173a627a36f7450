"""Auto-SPMD on a tiny GPT: trace/solve/transform/run, golden vs vanilla."""
import copy

import pytest
import torch

from easydist_amd.utils.testing import init_single_process, spawn


def _tiny_cfg():
    from easydist_amd.models.gpt import GPTConfig
    return GPTConfig(vocab_size=128, n_layer=2, n_head=2, n_embd=32,
                     block_size=32)


def _run_gpt_golden(ws):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
    from easydist_amd.models.gpt import GPT

    easydist_setup(device="cpu")
    set_device_mesh(list(range(ws)), ["spmd0"])
    torch.manual_seed(3)
    model = GPT(_tiny_cfg())
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3, fused=True)

    def train_step(model, opt, idx, targets):
        loss = model.loss(idx, targets)
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    compiled = easydist_compile(train_step, cuda_graph=False)

    torch.manual_seed(11)
    for step in range(3):
        idx = torch.randint(0, 128, (4, 32))
        tg = torch.randint(0, 128, (4, 32))
        dist.broadcast(idx, src=0)
        dist.broadcast(tg, src=0)
        loss = compiled(model, opt, idx, tg)
        ref = train_step(model_ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref)) < 1e-3, (step, float(loss),
                                                      float(ref))
    final = compiled.named_parameters()
    for n, p_ref in model_ref.named_parameters():
        assert torch.allclose(final[n], p_ref.detach(), rtol=1e-3,
                              atol=1e-4), n


def test_gpt_ws1():
    init_single_process()
    _run_gpt_golden(1)


@pytest.mark.world2
def test_gpt_ws2():
    spawn(_run_gpt_golden, args=(2,), world_size=2, port=29537)


def test_gpt_autocast_cpu_compile():
    """The bench train step (autocast bf16 + fused CE + lowered norms)
    through the auto pipeline on CPU — locks in the aten fallbacks of
    every lowered kernel."""
    import torch

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models import gpt as gptm
    from easydist_amd.utils.testing import init_single_process

    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])
    torch.manual_seed(0)
    cfg = gptm.GPTConfig(vocab_size=128, n_layer=2, n_head=2, n_embd=32,
                         block_size=16)
    model = gptm.GPT(cfg)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    compiled = easydist_compile(gptm.gpt_train_step, cuda_graph=False)
    idx = torch.randint(0, 128, (4, 16))
    tg = torch.randint(0, 128, (4, 16))
    losses = [float(compiled(model, opt, idx, tg)) for _ in range(4)]
    assert all(map(lambda x: x == x, losses)), losses   # no NaNs
    assert losses[-1] < losses[0], losses


def _gpt_ws4_body(world_size):
    """ws4 auto-SPMD on a mid-size GPT: large enough that the solver
    actually reshards (beam strategy, comm in the bwd), exercising the
    dtype-tolerant LN backward path; golden vs vanilla."""
    import copy
    from dataclasses import replace

    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.gpt import GPT, GPT2_SMALL, gpt_train_step

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = replace(GPT2_SMALL, n_layer=4, block_size=128)
    model = GPT(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-4,
                               fused=True)
    compiled = easydist_compile(gpt_train_step, cuda_graph=False)
    torch.manual_seed(5)
    for i in range(2):
        idx = torch.randint(0, cfg.vocab_size, (8, 128))
        tg = torch.randint(0, cfg.vocab_size, (8, 128))
        dist.broadcast(idx, src=0)
        dist.broadcast(tg, src=0)
        loss = compiled(model, opt, idx, tg)
        ref = gpt_train_step(model_ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref)) < 2e-2, \
            (i, float(loss), float(ref))


@pytest.mark.world4
def test_gpt_auto_ws4_midsize():
    spawn(_gpt_ws4_body, args=(4,), world_size=4, port=29638)


def _gpt_ws8_body(world_size):
    """ws8 (the driver's largest bench N) on a 2-layer GPT-2-small
    geometry: golden vs vanilla."""
    import copy
    from dataclasses import replace

    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.gpt import GPT, GPT2_SMALL, gpt_train_step

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = replace(GPT2_SMALL, n_layer=2, block_size=128)
    model = GPT(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-4,
                               fused=True)
    compiled = easydist_compile(gpt_train_step, cuda_graph=False)
    torch.manual_seed(5)
    for i in range(2):
        idx = torch.randint(0, cfg.vocab_size, (8, 128))
        tg = torch.randint(0, cfg.vocab_size, (8, 128))
        dist.broadcast(idx, src=0)
        dist.broadcast(tg, src=0)
        loss = compiled(model, opt, idx, tg)
        ref = gpt_train_step(model_ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref)) < 2e-2, \
            (i, float(loss), float(ref))


@pytest.mark.world8
def test_gpt_auto_ws8():
    spawn(_gpt_ws8_body, args=(8,), world_size=8, port=29648,
          timeout=900.0)
