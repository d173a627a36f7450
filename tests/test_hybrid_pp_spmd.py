"""Hybrid pipeline x SPMD golden test (CPU, gloo ws4: pp2 x spmd2).

reference: tests/test_torch/test_hybrid.py (mesh ['pp','spmd']): the
auto-SPMD pass shards each pipeline stage's graphs along the spmd mesh
dim; losses and params must match vanilla single-process training.
"""
import copy

import pytest
import torch
import torch.nn as nn

from easydist_amd.utils.testing import spawn


class MLP4(nn.Module):
    def __init__(self, d=16, h=32):
        super().__init__()
        self.fc1 = nn.Linear(d, h)
        self.relu = nn.ReLU()
        self.fc2 = nn.Linear(h, h)
        self.fc3 = nn.Linear(h, d)

    def forward(self, x):
        return self.fc3(self.fc2(self.relu(self.fc1(x))))


def train_step(model, opt, x, y):
    loss = ((model(x) - y) ** 2).mean()
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss


def _body(world_size):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([[0, 1], [2, 3]], ["pp", "spmd0"])

    torch.manual_seed(42)
    model = MLP4()
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-2)

    compiled = easydist_compile(train_step, parallel_mode="pp",
                                cuda_graph=False, split_points={"fc2"},
                                nchunks=2)
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


@pytest.mark.world4
def test_hybrid_pp2_spmd2():
    spawn(_body, args=(4,), world_size=4, port=29591)


def _gpt_body(world_size):
    """Hybrid pp2 x spmd2 on the real GPT model (flash-attention custom
    ops, fused CE, embeddings) — losses track vanilla training."""
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.gpt import GPT, GPTConfig

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([[0, 1], [2, 3]], ["pp", "spmd0"])

    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=64, n_layer=2, n_head=2, n_embd=32,
                    block_size=16)
    model = GPT(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3)

    def step(model, opt, idx, tg):
        loss = model.loss(idx, tg)
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    compiled = easydist_compile(step, parallel_mode="pp",
                                cuda_graph=False, split_points={"h.1"},
                                nchunks=2)
    torch.manual_seed(5)
    for i in range(3):
        idx = torch.randint(0, 64, (4, 16))
        tg = torch.randint(0, 64, (4, 16))
        dist.broadcast(idx, src=0)
        dist.broadcast(tg, src=0)
        loss = compiled(model, opt, idx, tg)
        ref = step(model_ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref)) < 5e-3, \
            (i, float(loss), float(ref))


@pytest.mark.world4
def test_hybrid_gpt_pp2_spmd2():
    spawn(_gpt_body, args=(4,), world_size=4, port=29624)
