"""Checkpoint/resume through the auto-SPMD runtime (CPU, ws1):
state_dict after K steps -> fresh compile -> load_state_dict -> the
continued trajectory must match an uninterrupted run. Also covers
compiling against an optimizer that ALREADY has state (resume from an
eager phase): the trace warmup must not corrupt live moments."""
import copy

import torch
import torch.nn as nn

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
from easydist_amd.utils.testing import init_single_process


class _Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.a = nn.Linear(12, 24)
        self.b = nn.Linear(24, 12)

    def forward(self, x):
        return self.b(torch.tanh(self.a(x)))


def _step(model, opt, x, y):
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss


def _data(n):
    torch.manual_seed(100)
    return [(torch.randn(4, 12), torch.randn(4, 12)) for _ in range(n)]


def _setup():
    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])


def test_checkpoint_resume_roundtrip():
    _setup()
    torch.manual_seed(0)
    model = _Net()
    model_b = copy.deepcopy(model)
    data = _data(4)

    # uninterrupted 4 steps
    opt = torch.optim.Adam(model.parameters(), lr=1e-2, fused=True)
    ca = easydist_compile(_step, cuda_graph=False)
    for x, y in data:
        ca(model, opt, x, y)
    want_params = dict(ca.named_parameters())

    # 2 steps -> checkpoint -> fresh everything -> load -> 2 more steps
    opt_b = torch.optim.Adam(model_b.parameters(), lr=1e-2, fused=True)
    cb = easydist_compile(_step, cuda_graph=False)
    for x, y in data[:2]:
        cb(model_b, opt_b, x, y)
    ckpt = list(cb.compiled.values())[0].state_dict()

    torch.manual_seed(42)
    model_c = _Net()
    opt_c = torch.optim.Adam(model_c.parameters(), lr=1e-2, fused=True)
    cc = easydist_compile(_step, cuda_graph=False)
    # one dummy call compiles + seeds internal state; then overwrite it
    cc(model_c, opt_c, *data[0])
    list(cc.compiled.values())[0].load_state_dict(ckpt)
    for x, y in data[2:]:
        cc(model_c, opt_c, x, y)

    # compare against the uninterrupted run's params
    resumed = dict(cc.named_parameters())
    for n, p in want_params.items():
        assert torch.allclose(resumed[n], p, rtol=1e-5, atol=1e-6), \
            (n, (resumed[n] - p).abs().max())


def test_compile_with_prestep_optimizer():
    """Eager-train 2 steps, then hand the live model+optimizer to
    easydist_compile: the compile-time warmup must preserve the existing
    moments and the continued trajectory must match pure eager."""
    _setup()
    torch.manual_seed(1)
    model = _Net()
    model_ref = copy.deepcopy(model)
    mk = lambda m: torch.optim.Adam(m.parameters(), lr=1e-2,
                                    weight_decay=0.01, fused=True)
    opt, opt_ref = mk(model), mk(model_ref)
    data = _data(5)

    for x, y in data[:2]:            # eager phase (both)
        _step(model, opt, x, y)
        _step(model_ref, opt_ref, x, y)

    compiled = easydist_compile(_step, cuda_graph=False)
    for i, (x, y) in enumerate(data[2:]):
        loss = compiled(model, opt, x, y)
        ref = _step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref)) < 1e-5, \
            (i, float(loss), float(ref))
    live = dict(compiled.named_parameters())
    for n, pr in model_ref.named_parameters():
        assert torch.allclose(live[n], pr, rtol=1e-5, atol=1e-6), \
            (n, (live[n] - pr).abs().max())
