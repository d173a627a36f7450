"""Ring attention (sequence parallel) golden tests: exact match vs full
attention on the gathered sequence, forward AND backward (CPU, gloo)."""
import math

import pytest
import torch

from easydist_amd.utils.testing import spawn


def _full_ref(q, k, v, causal):
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal:
        S = s.shape[-1]
        mask = torch.ones(S, S, dtype=torch.bool).tril()
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, v.float())


def _body(world_size, causal):
    import torch.distributed as dist

    from easydist_amd.ops.ring_attention import ring_attention

    B, H, S, D = 2, 3, 16, 8
    r = dist.get_rank()
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    for t in (q, k, v):
        dist.broadcast(t, src=0)
    qf, kf, vf = (t.clone().requires_grad_(True) for t in (q, k, v))
    ref = _full_ref(qf, kf, vf, causal)
    g = torch.randn_like(ref)
    dist.broadcast(g, src=0)
    ref.backward(g)

    Sl = S // world_size
    sl = slice(r * Sl, (r + 1) * Sl)
    ql = q[:, :, sl].clone().requires_grad_(True)
    kl = k[:, :, sl].clone().requires_grad_(True)
    vl = v[:, :, sl].clone().requires_grad_(True)
    out = ring_attention(ql, kl, vl, group=dist.group.WORLD, causal=causal)
    assert torch.allclose(out.float(), ref.detach()[:, :, sl], rtol=1e-4,
                          atol=1e-5), (out.float()
                                       - ref.detach()[:, :, sl]).abs().max()
    out.backward(g[:, :, sl])
    for got, want, name in ((ql.grad, qf.grad[:, :, sl], "dq"),
                            (kl.grad, kf.grad[:, :, sl], "dk"),
                            (vl.grad, vf.grad[:, :, sl], "dv")):
        assert torch.allclose(got.float(), want, rtol=1e-4, atol=1e-5), \
            (name, (got.float() - want).abs().max())


@pytest.mark.world2
@pytest.mark.parametrize("causal", [True, False])
def test_ring_attention_ws2(causal):
    spawn(_body, args=(2, causal), world_size=2, port=29581 + int(causal))


@pytest.mark.world4
def test_ring_attention_ws4_causal():
    spawn(_body, args=(4, True), world_size=4, port=29585)


def _gpt_sp_body(world_size):
    """Full GPT fwd+bwd with the sequence sharded across ranks: loss and
    wte grads must match the single-sequence run."""
    import torch.distributed as dist

    from easydist_amd.models.gpt import GPT, GPTConfig

    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=64, n_layer=2, n_head=2, n_embd=32,
                    block_size=32)
    model = GPT(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    import copy
    model_ref = copy.deepcopy(model)
    model.enable_sequence_parallel(dist.group.WORLD)

    torch.manual_seed(3)
    idx = torch.randint(0, 64, (2, 32))
    tg = torch.randint(0, 64, (2, 32))
    dist.broadcast(idx, src=0)
    dist.broadcast(tg, src=0)
    r = dist.get_rank()
    Sl = 32 // world_size
    sl = slice(r * Sl, (r + 1) * Sl)

    # local CE SUM over my token shard; global mean = all-reduced sum / N
    logits = model(idx[:, sl])
    lsum = torch.nn.functional.cross_entropy(
        logits.reshape(-1, 64), tg[:, sl].reshape(-1), reduction="sum")
    loss = lsum / tg.numel()
    loss.backward()
    total = loss.detach().clone()
    dist.all_reduce(total)

    ref_loss = torch.nn.functional.cross_entropy(
        model_ref(idx).reshape(-1, 64), tg.reshape(-1))
    ref_loss.backward()
    assert abs(float(total) - float(ref_loss)) < 1e-5, \
        (float(total), float(ref_loss))
    # wte grad: local shards contribute disjoint/overlapping token rows;
    # all-reduce reconstructs the full grad
    g = model.wte.weight.grad.clone()
    dist.all_reduce(g)
    assert torch.allclose(g, model_ref.wte.weight.grad, rtol=1e-4,
                          atol=1e-5), (g - model_ref.wte.weight.grad
                                       ).abs().max()


@pytest.mark.world2
def test_gpt_sequence_parallel_ws2():
    spawn(_gpt_sp_body, args=(2,), world_size=2, port=29587)
