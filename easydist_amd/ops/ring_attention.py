"""Ring attention: sequence-parallel exact attention over the xGMI ring.

The reference has NO long-context strategy (no ring
attention, no Ulysses, no context parallel) and BASELINE's north star
requires one. MI355X-first design:

* the sequence is sharded S -> W x S/W across an 'sp' mesh dimension;
  each ring step overlaps one neighbor KV-block hop (7 xGMI p2p links
  make neighbor exchange ~free relative to the block attention GEMMs)
  with the local block-attention compute;
* per-block attention reuses the gfx950 flash kernel (or the fp32 math
  path on CPU); partial results merge by log-sum-exp accumulation, so
  the result is EXACT attention, not an approximation;
* causal masking with a sequence-ordered shard layout: KV blocks from
  earlier ranks attend fully, the own block causally, later blocks are
  skipped (their hop still happens to keep the ring in lockstep);
* backward is the standard two-ring recompute: ring 1 accumulates dQ
  locally, dK/dV accumulate into a rotating buffer that arrives back at
  its owner after W hops.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.distributed as dist


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


def _block_attn(q, k, v, causal_mode: str):
    """One block attention returning (out, lse). causal_mode:
    'full' | 'causal' | 'skip'."""
    if causal_mode == "skip":
        out = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        lse = torch.full(q.shape[:-1], float("-inf"), dtype=torch.float32,
                         device=q.device)
        return out, lse
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal_mode == "causal":
        S, T = s.shape[-2], s.shape[-1]
        mask = torch.ones(S, T, dtype=torch.bool, device=s.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.exp(s - lse.unsqueeze(-1))
    p = torch.nan_to_num(p, nan=0.0)        # rows fully masked
    out = torch.matmul(p, v.float())
    return out, lse


def _merge(out, lse, out_b, lse_b):
    """Log-sum-exp merge of two partial attention results."""
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
    kv = torch.stack([k.float(), v.float()])
    for step in range(1, w):
        kv = _ring_exchange(kv, group)
        src = (rank - step) % w
        mode = _mode(src, rank, causal)
        if mode != "skip":
            out_b, lse_b = _block_attn(q, kv[0].to(q.dtype),
                                       kv[1].to(q.dtype), mode)
            out, lse = _merge(out, lse, out_b, lse_b)
        # skipped blocks still traveled: ring stays in lockstep
    return out.to(q.dtype), lse


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
        group, causal = ctx.group, ctx.causal
        w = dist.get_world_size(group)
        rank = dist.get_rank(group)
        scale = 1.0 / math.sqrt(q.shape[-1])
        gradf = grad.float()
        qf = q.float()
        # delta = rowsum(dO * O) — constant across KV blocks
        delta = (gradf * out.float()).sum(-1, keepdim=True)

        dq = torch.zeros_like(qf)
        # rotating buffer: [k, v, dk, dv] — dk/dv accumulate as the block
        # passes by and arrive home after the final hop
        buf = torch.stack([k.float(), v.float(),
                           torch.zeros_like(k, dtype=torch.float32),
                           torch.zeros_like(v, dtype=torch.float32)])
        with torch.no_grad():
            for step in range(w):
                if step > 0:
                    buf = _ring_exchange(buf, group)
                src = (rank - step) % w
                mode = _mode(src, rank, causal)
                if mode != "skip":
                    kf, vf = buf[0], buf[1]
                    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
                    if mode == "causal":
                        S, T = s.shape[-2], s.shape[-1]
                        m = torch.ones(S, T, dtype=torch.bool,
                                       device=s.device).tril()
                        s = s.masked_fill(~m, float("-inf"))
                    p = torch.exp(s - lse.float().unsqueeze(-1))
                    p = torch.nan_to_num(p, nan=0.0)
                    dv_b = torch.matmul(p.transpose(-1, -2), gradf)
                    dp = torch.matmul(gradf, vf.transpose(-1, -2))
                    ds = p * (dp - delta) * scale
                    dq += torch.matmul(ds, kf)
                    buf[2] += torch.matmul(ds.transpose(-1, -2), qf)
                    buf[3] += dv_b
            # one final hop returns each block (with its grads) home
            buf = _ring_exchange(buf, group)
        return (dq.to(q.dtype), buf[2].to(k.dtype), buf[3].to(v.dtype),
                None, None)


def ring_attention(q, k, v, group=None, causal: bool = True):
    """Exact sequence-parallel attention; q,k,v: [B, H, S_local, D]."""
    if group is None or dist.get_world_size(group) == 1:
        from .attention import scaled_dot_product_attention
        return scaled_dot_product_attention(q, k, v, causal=causal)
    return _RingAttention.apply(q, k, v, group, causal)
