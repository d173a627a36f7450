"""Ring attention: sequence-parallel exact attention over the xGMI ring.

The reference has NO long-context strategy (no ring
attention, no Ulysses, no context parallel) and BASELINE's north star
requires one. MI355X-first design:

* the sequence is sharded S -> W x S/W across an 'sp' mesh dimension;
  each ring step overlaps one neighbor KV-block hop (7 xGMI p2p links
  make neighbor exchange ~free relative to the block attention GEMMs)
  with the local block-attention compute;
* per-block attention goes through the ``easydist_amd::flash_attention``
  custom op — the gfx950 flash kernel on GPU (it returns the LSE the
  merge needs), the fp32 math path on CPU; partial results merge by
  log-sum-exp accumulation in fp32, so the result is EXACT attention,
  not an approximation;
* ring buffers travel in the SOURCE dtype (bf16 K/V hop = half the xGMI
  traffic of an fp32 hop); only the dK/dV accumulators ride fp32;
* causal masking with a sequence-ordered shard layout: KV blocks from
  earlier ranks attend fully, the own block causally, later blocks are
  skipped (their hop still happens to keep the ring in lockstep);
* backward is the standard two-ring recompute: ring 1 accumulates dQ
  locally, dK/dV accumulate into a rotating fp32 buffer that arrives
  back at its owner after W hops.
"""
from __future__ import annotations

from typing import Tuple

import torch
import torch.distributed as dist

from . import attention as _attention  # noqa: F401  (registers the ops)


def _ring_exchange(t: torch.Tensor, group) -> torch.Tensor:
    """Send my tensor to rank+1 (mod W), receive from rank-1 (mod W)."""
    w = dist.get_world_size(group)
    r = dist.get_rank(group)
    ranks = dist.get_process_group_ranks(group) if group is not None \
        else list(range(w))
    nxt = ranks[(r + 1) % w]
    prv = ranks[(r - 1) % w]
    recv = torch.empty_like(t)
    t = t.contiguous()
    ops = [dist.P2POp(dist.isend, t, nxt, group=group),
           dist.P2POp(dist.irecv, recv, prv, group=group)]
    for wk in dist.batch_isend_irecv(ops):
        wk.wait()
    return recv


def _exchange_pair(a: torch.Tensor, b: torch.Tensor, group
                   ) -> Tuple[torch.Tensor, torch.Tensor]:
    """One ring hop of two same-dtype tensors as a single message."""
    st = _ring_exchange(torch.stack([a, b]), group)
    return st[0], st[1]


def _block_attn(q, k, v, causal_mode: str):
    """One block attention returning (out_f32, lse_f32). causal_mode:
    'full' | 'causal' (callers handle 'skip')."""
    out, lse = torch.ops.easydist_amd.flash_attention(
        q, k, v, causal_mode == "causal")
    return out.float(), lse.float()


def _merge(out, lse, out_b, lse_b):
    """Log-sum-exp merge of two partial attention results (fp32)."""
    new_lse = torch.logaddexp(lse, lse_b)
    a = torch.exp(lse - new_lse).unsqueeze(-1)
    b = torch.exp(lse_b - new_lse).unsqueeze(-1)
    return out * a + out_b * b, new_lse


def _mode(src: int, rank: int, causal: bool) -> str:
    if not causal:
        return "full"
    if src < rank:
        return "full"
    if src == rank:
        return "causal"
    return "skip"


def _ring_forward(q, k, v, group, causal):
    w = dist.get_world_size(group)
    rank = dist.get_rank(group)
    out, lse = _block_attn(q, k, v, "causal" if causal else "full")
    kb, vb = k, v
    for step in range(1, w):
        kb, vb = _exchange_pair(kb, vb, group)   # source dtype on the wire
        src = (rank - step) % w
        mode = _mode(src, rank, causal)
        if mode != "skip":
            out_b, lse_b = _block_attn(q, kb, vb, mode)
            out, lse = _merge(out, lse, out_b, lse_b)
        # skipped blocks still traveled: ring stays in lockstep
    return out.to(q.dtype), lse


def ring_bwd(grad, q, k, v, out, lse, group, causal):
    """Two-ring exact backward; the per-block math is exactly
    flash_attention_bwd evaluated against the GLOBAL out/lse."""
    w = dist.get_world_size(group)
    rank = dist.get_rank(group)
    dq = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
    kb, vb = k, v                       # rotating K/V, source dtype
    dkv = torch.zeros((2,) + k.shape, dtype=torch.float32,
                      device=k.device)  # rotating dK/dV, fp32
    grad = grad.contiguous()
    with torch.no_grad():
        for step in range(w):
            if step > 0:
                kb, vb = _exchange_pair(kb, vb, group)
                dkv = _ring_exchange(dkv, group)
            src = (rank - step) % w
            mode = _mode(src, rank, causal)
            if mode != "skip":
                dq_b, dk_b, dv_b = torch.ops.easydist_amd.\
                    flash_attention_bwd(grad, q, kb, vb, out, lse,
                                        mode == "causal")
                dq += dq_b.float()
                dkv[0] += dk_b.float()
                dkv[1] += dv_b.float()
        # one final hop returns each block (with its grads) home
        dkv = _ring_exchange(dkv, group)
    return dq.to(q.dtype), dkv[0].to(k.dtype), dkv[1].to(v.dtype)


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal):
        with torch.no_grad():
            out, lse = _ring_forward(q.detach(), k.detach(), v.detach(),
                                     group, causal)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group = group
        ctx.causal = causal
        return out

    @staticmethod
    def backward(ctx, grad):
        q, k, v, out, lse = ctx.saved_tensors
        dq, dk, dv = ring_bwd(grad, q, k, v, out, lse, ctx.group,
                              ctx.causal)
        return dq, dk, dv, None, None


def ring_attention(q, k, v, group=None, causal: bool = True):
    """Exact sequence-parallel attention; q,k,v: [B, H, S_local, D]."""
    if group is None or dist.get_world_size(group) == 1:
        from .attention import scaled_dot_product_attention
        return scaled_dot_product_attention(q, k, v, causal=causal)
    return _RingAttention.apply(q, k, v, group, causal)
