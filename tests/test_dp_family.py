"""Golden tests for the manual DP family: ddp / zero2 / zero3 (CPU, gloo).

Mirrors the reference's golden pattern (tests/test_torch/test_spmd.py
60-117) applied to compile_dp (reference easydist/torch/compile_dp.py):
vanilla torch on the global batch vs the transformed graph on per-rank
shards; loss, params and optimizer states must match.
"""
import copy

import pytest
import torch
import torch.nn as nn

from easydist_amd.utils.testing import init_single_process, spawn


class MLP(nn.Module):
    def __init__(self, d=16, h=37):   # h=37: odd so FLAT padding is hit
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


def _run_golden(world_size, mode):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])

    torch.manual_seed(42)
    model = MLP()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)

    opt = torch.optim.Adam(model.parameters(), lr=1e-2, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-2, fused=True)

    compiled = easydist_compile(train_step, parallel_mode=mode,
                                cuda_graph=False)

    torch.manual_seed(7)
    for step in range(4):
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


@pytest.mark.parametrize("mode", ["ddp", "zero2", "zero3"])
def test_dp_ws1(mode):
    init_single_process()
    _run_golden(1, mode)


@pytest.mark.world2
@pytest.mark.parametrize("mode", ["ddp", "zero2", "zero3"])
def test_dp_ws2(mode):
    spawn(_run_golden, args=(2, mode), world_size=2, port=29536)


def _gpt_zero_body(world_size, mode):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.gpt import GPT, GPTConfig

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=128, n_layer=2, n_head=2, n_embd=32,
                    block_size=16)
    model = GPT(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3, fused=True)

    def step(model, opt, idx, tg):
        loss = model.loss(idx, tg)
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    compiled = easydist_compile(step, parallel_mode=mode, cuda_graph=False)
    torch.manual_seed(3)
    for i in range(3):
        idx = torch.randint(0, 128, (4, 16))
        tg = torch.randint(0, 128, (4, 16))
        dist.broadcast(idx, src=0)
        dist.broadcast(tg, src=0)
        loss = compiled(model, opt, idx, tg)
        ref = step(model_ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref)) < 5e-4, \
            (i, float(loss), float(ref))


@pytest.mark.world2
@pytest.mark.parametrize("mode", ["zero2", "zero3"])
def test_gpt_zero_ws2(mode):
    spawn(_gpt_zero_body, args=(2, mode), world_size=2,
          port=29547 + (mode == "zero3"))


def _ddp_sgd_body(ws):
    """ddp mode with the fused SGD optimizer (momentum): grads average
    across ranks, trajectory matches single-process full-batch SGD."""
    import copy

    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(ws)), ["spmd0"])
    torch.manual_seed(0)
    model = MLP()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.SGD(model.parameters(), lr=1e-2, momentum=0.9,
                          fused=True)
    opt_ref = torch.optim.SGD(model_ref.parameters(), lr=1e-2,
                              momentum=0.9, fused=True)
    compiled = easydist_compile(train_step, parallel_mode="ddp",
                                cuda_graph=False)
    torch.manual_seed(7)
    for i in range(3):
        # dp contract: every rank passes the GLOBAL batch; the compiled
        # wrapper chunks it per rank and all-reduce(avg)s grads + loss
        x = torch.randn(8, 16)
        y = torch.randn(8, 16)
        dist.broadcast(x, src=0)
        dist.broadcast(y, src=0)
        loss = compiled(model, opt, x, y)
        ref = train_step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref)) < 1e-5, \
            (i, float(loss), float(ref))


@pytest.mark.world2
def test_ddp_sgd_momentum_ws2():
    spawn(_ddp_sgd_body, args=(2,), world_size=2, port=29642)


def _zero2_adamw_body(ws):
    """zero2 with AdamW: FLAT-sharded moments must follow the decoupled-
    decay math; golden vs single-process full batch."""
    import copy

    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(ws)), ["spmd0"])
    torch.manual_seed(0)
    model = MLP()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-2, weight_decay=0.05,
                            fused=True)
    opt_ref = torch.optim.AdamW(model_ref.parameters(), lr=1e-2,
                                weight_decay=0.05, fused=True)
    compiled = easydist_compile(train_step, parallel_mode="zero2",
                                cuda_graph=False)
    torch.manual_seed(7)
    for i in range(3):
        x = torch.randn(8, 16)
        y = torch.randn(8, 16)
        dist.broadcast(x, src=0)
        dist.broadcast(y, src=0)
        loss = compiled(model, opt, x, y)
        ref = train_step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref)) < 1e-5, \
            (i, float(loss), float(ref))


@pytest.mark.world2
def test_zero2_adamw_ws2():
    spawn(_zero2_adamw_body, args=(2,), world_size=2, port=29656)
