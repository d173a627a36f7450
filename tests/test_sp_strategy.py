"""Sequence parallelism as a SOLVER strategy.

At real long-context scale (S=32k, single-head) the solver must assign
S(seq) to the flash-attention inputs — the only way to parallelize the
quadratic term — and the sharding transform rewrites the node to the
ring-attention runtime. The solve is asserted WITHOUT executing (a 32k
math-path attention materializes S^2 scores on CPU); the ring runtime
itself is golden-tested directly against full attention below and in
tests/test_ring_attention.py.
"""
import copy
from dataclasses import replace

import pytest
import torch

from easydist_amd.utils.testing import spawn


def _solver_body(world_size):
    import torch.distributed as dist

    from easydist_amd import easydist_setup, set_device_mesh
    from easydist_amd.compiler.compile_auto import shard_graph
    from easydist_amd.compiler.passes.functionalize import canonicalize
    from easydist_amd.compiler.tracing import ed_compile_func
    from easydist_amd.models import gpt as gptm
    from easydist_amd.parallel.device_mesh import get_device_mesh

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = replace(gptm.GPT2_SMALL, n_layer=1, n_embd=64, n_head=1,
                  block_size=32768, vocab_size=512)
    model = gptm.GPT(cfg)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, foreach=False)

    def train_step(model, opt, idx, targets):
        return gptm.gpt_train_step(model, opt, idx, targets)

    idx = torch.randint(0, 512, (1, 32768))
    tg = torch.randint(0, 512, (1, 32768))
    params, buffers, named_states, gm = ed_compile_func(
        train_step, "fake", (model, opt, idx, tg), {}, model, opt)
    gm, io_map = canonicalize(gm)
    gm2, env, _, _ = shard_graph(gm, get_device_mesh(), io_map, set(), "cpu")
    names = [getattr(n.target, "__name__", "") for n in gm2.graph.nodes
             if n.op == "call_function"]
    assert "rt_ring_attention" in names, \
        [s for s in names if "attention" in s or s.startswith("rt_")]
    assert "rt_ring_attention_bwd" in names


@pytest.mark.world2
def test_sp_solver_chooses_ring_ws2():
    spawn(_solver_body, args=(2,), world_size=2, port=29566)


def _runtime_body(world_size):
    """rt_ring_attention(+bwd) against full-sequence attention."""
    import math

    import torch.distributed as dist

    from easydist_amd import easydist_setup, set_device_mesh
    from easydist_amd.runtime import comm_runtime as crt

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    r = dist.get_rank()

    B, H, S, D = 2, 2, 32, 8
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    g = torch.randn(B, H, S, D)
    for t in (q, k, v, g):
        dist.broadcast(t, src=0)
    # full reference
    qf = q.clone().requires_grad_(True)
    kf = k.clone().requires_grad_(True)
    vf = v.clone().requires_grad_(True)
    scale = 1.0 / math.sqrt(D)
    s = torch.matmul(qf.float(), kf.float().transpose(-1, -2)) * scale
    mask = torch.ones(S, S, dtype=torch.bool).tril()
    s = s.masked_fill(~mask, float("-inf"))
    ref = torch.matmul(torch.softmax(s, -1), vf.float())
    ref.backward(g.float())

    Sl = S // world_size
    sl = slice(r * Sl, (r + 1) * Sl)
    ql, kl, vl = (t[:, :, sl].contiguous() for t in (q, k, v))
    out, lse = crt.rt_ring_attention(ql, kl, vl, True, 0)
    assert torch.allclose(out.float(), ref.detach()[:, :, sl], rtol=1e-4,
                          atol=1e-5)
    dq, dk, dv = crt.rt_ring_attention_bwd(g[:, :, sl].contiguous(), ql, kl,
                                           vl, out.to(q.dtype), lse, True, 0)
    for got, want in ((dq, qf.grad), (dk, kf.grad), (dv, vf.grad)):
        assert torch.allclose(got.float(), want[:, :, sl], rtol=1e-4,
                              atol=1e-5), float(
            (got.float() - want[:, :, sl]).abs().max())


@pytest.mark.world2
def test_rt_ring_attention_ws2():
    spawn(_runtime_body, args=(2,), world_size=2, port=29567)
