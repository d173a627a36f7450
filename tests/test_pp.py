"""Pipeline parallelism golden tests (CPU; local mode + gloo ws2).

Mirrors the reference's PP test shape (tests/test_torch/test_pp/
test_runtime.py: GPipe & DAPPLE x {MLP, GPT} x split annotations, golden
vs vanilla; test_split.py: local single-process stage execution).
"""
import copy

import pytest
import torch
import torch.nn as nn

from easydist_amd.utils.testing import init_single_process, spawn


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


def _golden_loop(model_ctor, opt_ctor, compiled_kwargs, steps=4, d=16):
    from easydist_amd import easydist_compile

    torch.manual_seed(42)
    model = model_ctor()
    model_ref = copy.deepcopy(model)
    opt = opt_ctor(model)
    opt_ref = opt_ctor(model_ref)
    compiled = easydist_compile(train_step, parallel_mode="pp",
                                cuda_graph=False, **compiled_kwargs)
    torch.manual_seed(7)
    for step in range(steps):
        x = torch.randn(8, d)
        y = torch.randn(8, d)
        loss = compiled(model, opt, x, y)
        ref = train_step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref)) < 1e-4, \
            (step, float(loss), float(ref))
    final = compiled.named_parameters()
    for n, p_ref in model_ref.named_parameters():
        assert torch.allclose(final[n], p_ref.detach(), rtol=1e-4,
                              atol=1e-5), (n,)


def test_pp_local_split_points():
    init_single_process()
    _golden_loop(MLP4, lambda m: torch.optim.Adam(m.parameters(), lr=1e-2),
                 dict(split_points={"fc2"}, nchunks=4))


def test_pp_local_equal_size():
    init_single_process()
    _golden_loop(MLP4, lambda m: torch.optim.Adam(m.parameters(), lr=1e-2),
                 dict(nstages=2, nchunks=2))


def test_pp_local_sgd():
    init_single_process()
    _golden_loop(MLP4, lambda m: torch.optim.SGD(m.parameters(), lr=1e-2,
                                                 momentum=0.9),
                 dict(split_points={"fc2"}, nchunks=4))


def _dist_body(world_size, schedule):
    _golden_loop(MLP4, lambda m: torch.optim.Adam(m.parameters(), lr=1e-2),
                 dict(split_points={"fc2"}, nchunks=4, schedule=schedule))


@pytest.mark.world2
@pytest.mark.parametrize("schedule", ["gpipe", "dapple"])
def test_pp_ws2(schedule):
    spawn(_dist_body, args=(2, schedule), world_size=2,
          port=29541 + (schedule == "dapple"))


def _gpt_body(world_size, schedule):
    from easydist_amd import easydist_compile
    from easydist_amd.models.gpt import GPT, GPTConfig

    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=128, n_layer=4, n_head=2, n_embd=32,
                    block_size=16)
    model = GPT(cfg)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3)

    def gpt_step(model, opt, idx, tg):
        loss = model.loss(idx, tg)
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    compiled = easydist_compile(gpt_step, parallel_mode="pp",
                                cuda_graph=False,
                                split_points={"h.1"}, nchunks=2,
                                schedule=schedule)
    torch.manual_seed(5)
    for step in range(3):
        idx = torch.randint(0, 128, (4, 16))
        tg = torch.randint(0, 128, (4, 16))
        loss = compiled(model, opt, idx, tg)
        ref = gpt_step(model_ref, opt_ref, idx, tg)
        assert abs(float(loss) - float(ref)) < 5e-4, \
            (step, float(loss), float(ref))


def test_pp_gpt_local():
    init_single_process()
    _gpt_body(1, "gpipe")


@pytest.mark.world2
def test_pp_gpt_ws2_dapple():
    spawn(_gpt_body, args=(2, "dapple"), world_size=2, port=29545)


def test_pp_state_dict_roundtrip():
    """Checkpoint parity: save at step 2, train to 4, load, rerun — losses
    must repeat exactly (reference: pp/runtime.py:509-544)."""
    from easydist_amd import easydist_compile

    init_single_process()
    torch.manual_seed(42)
    model = MLP4()
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    compiled = easydist_compile(train_step, parallel_mode="pp",
                                cuda_graph=False, split_points={"fc2"},
                                nchunks=2)
    torch.manual_seed(7)
    batches = [(torch.randn(8, 16), torch.randn(8, 16)) for _ in range(4)]
    for x, y in batches[:2]:
        compiled(model, opt, x, y)
    rt = list(compiled.compiled.values())[0]
    ckpt = rt.state_dict()
    later = [float(compiled(model, opt, x, y)) for x, y in batches[2:]]
    rt.load_state_dict(ckpt)
    replay = [float(compiled(model, opt, x, y)) for x, y in batches[2:]]
    assert later == replay, (later, replay)


def test_pp_nchunks_edge_cases():
    """nchunks == 1 (no pipelining) and nchunks > nstages."""
    init_single_process()
    for nchunks in (1, 8):
        _golden_loop(MLP4,
                     lambda m: torch.optim.Adam(m.parameters(), lr=1e-2),
                     dict(split_points={"fc2"}, nchunks=nchunks), steps=2)


def test_pp_tied_weights_error_is_loud():
    """Tied weights spanning a split (BERT-style shared embedding/decoder)
    are not supported by pipeline mode: the compile must fail with an
    explanatory error instead of silently training wrong."""
    import pytest as _pytest

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.utils.testing import init_single_process

    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["pp"])

    class Tied(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb = nn.Embedding(16, 8)
            self.mid = nn.Linear(8, 8)
            self.head = nn.Linear(8, 16, bias=False)
            self.head.weight = self.emb.weight     # tied across the split

        def forward(self, idx):
            return self.head(torch.relu(self.mid(self.emb(idx))))

    model = Tied()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)

    def step(model, opt, idx, tg):
        loss = torch.nn.functional.cross_entropy(model(idx), tg)
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    compiled = easydist_compile(step, parallel_mode="pp",
                                cuda_graph=False, split_points={"mid"},
                                nchunks=1)
    idx = torch.randint(0, 16, (4,))
    tg = torch.randint(0, 16, (4,))
    with _pytest.raises(RuntimeError, match="multiple pipeline stages"):
        compiled(model, opt, idx, tg)
