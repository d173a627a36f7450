"""Manual tensor-parallel GPT golden tests (CPU, gloo): the TP model
loaded from a replicated GPT must match it exactly in fwd loss and in
sharded grads, mirroring the reference's manual-TP comparison baseline
(reference: benchmark/torch/model/gpt_tp.py)."""
import copy

import pytest
import torch

from easydist_amd.utils.testing import spawn


def _body(world_size):
    import torch.distributed as dist

    from easydist_amd.models.gpt import GPT, GPTConfig
    from easydist_amd.models.gpt_tp import GPT_TP

    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=64, n_layer=2, n_head=4, n_embd=32,
                    block_size=16)
    ref = GPT(cfg)
    for p in ref.parameters():
        dist.broadcast(p.data, src=0)

    tp = GPT_TP(cfg)
    tp.load_from_replicated(ref)

    torch.manual_seed(3)
    idx = torch.randint(0, 64, (2, 16))
    tg = torch.randint(0, 64, (2, 16))
    dist.broadcast(idx, src=0)
    dist.broadcast(tg, src=0)

    loss_tp = tp.loss(idx, tg)
    loss_ref = ref.loss(idx, tg)
    assert abs(float(loss_tp) - float(loss_ref)) < 1e-5, \
        (float(loss_tp), float(loss_ref))

    loss_tp.backward()
    loss_ref.backward()
    r = dist.get_rank()
    E = cfg.n_embd
    Ls = E // world_size
    H = 4 * E // world_size
    # replicated modules: grads must agree with the reference everywhere
    for name in ("wte", "wpe", "lm_head", "ln_f"):
        for (pn, p), (_, pr) in zip(
                getattr(tp, name).named_parameters(),
                getattr(ref, name).named_parameters()):
            assert torch.allclose(p.grad, pr.grad, rtol=1e-4, atol=1e-5), \
                (name, pn, (p.grad - pr.grad).abs().max())
    # sharded modules: grad shard == reference grad slice
    for blk, rblk in zip(tp.h, ref.h):
        qw, kw, vw = rblk.attn.c_attn.weight.grad.split(E, dim=0)
        want = torch.cat([t[r * Ls:(r + 1) * Ls] for t in (qw, kw, vw)], 0)
        got = blk.attn.c_attn.weight.grad
        assert torch.allclose(got, want, rtol=1e-4, atol=1e-5), \
            ("c_attn", (got - want).abs().max())
        got = blk.attn.c_proj.weight.grad
        want = rblk.attn.c_proj.weight.grad[:, r * Ls:(r + 1) * Ls]
        assert torch.allclose(got, want, rtol=1e-4, atol=1e-5), \
            ("attn.c_proj", (got - want).abs().max())
        got = blk.mlp.c_fc.weight.grad
        want = rblk.mlp.c_fc.weight.grad[r * H:(r + 1) * H]
        assert torch.allclose(got, want, rtol=1e-4, atol=1e-5), \
            ("c_fc", (got - want).abs().max())
        got = blk.mlp.c_proj.weight.grad
        want = rblk.mlp.c_proj.weight.grad[:, r * H:(r + 1) * H]
        assert torch.allclose(got, want, rtol=1e-4, atol=1e-5), \
            ("mlp.c_proj", (got - want).abs().max())
        # row-parallel bias is replicated
        if blk.mlp.c_proj.bias is not None:
            assert torch.allclose(blk.mlp.c_proj.bias.grad,
                                  rblk.mlp.c_proj.bias.grad,
                                  rtol=1e-4, atol=1e-5)


@pytest.mark.world2
def test_gpt_tp_ws2():
    spawn(_body, args=(2,), world_size=2, port=29621)


@pytest.mark.world4
def test_gpt_tp_ws4():
    spawn(_body, args=(4,), world_size=4, port=29622)


def _train_body(world_size):
    """3 Adam steps: TP training trajectory matches the replicated
    model's (losses to 1e-5)."""
    import torch.distributed as dist

    from easydist_amd.models.gpt import GPT, GPTConfig
    from easydist_amd.models.gpt_tp import GPT_TP

    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=64, n_layer=2, n_head=4, n_embd=32,
                    block_size=16)
    ref = GPT(cfg)
    for p in ref.parameters():
        dist.broadcast(p.data, src=0)
    tp = GPT_TP(cfg)
    tp.load_from_replicated(ref)
    opt = torch.optim.Adam(tp.parameters(), lr=1e-3)
    opt_ref = torch.optim.Adam(ref.parameters(), lr=1e-3)
    torch.manual_seed(5)
    for i in range(3):
        idx = torch.randint(0, 64, (2, 16))
        tg = torch.randint(0, 64, (2, 16))
        dist.broadcast(idx, src=0)
        dist.broadcast(tg, src=0)
        l = tp.loss(idx, tg)
        l.backward()
        opt.step(); opt.zero_grad(True)
        lr_ = ref.loss(idx, tg)
        lr_.backward()
        opt_ref.step(); opt_ref.zero_grad(True)
        assert abs(float(l) - float(lr_)) < 1e-4, (i, float(l), float(lr_))


@pytest.mark.world2
def test_gpt_tp_train_ws2():
    spawn(_train_body, args=(2,), world_size=2, port=29623)
