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
