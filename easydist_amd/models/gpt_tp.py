"""Manual tensor-parallel GPT — the hand-written comparison baseline.

Capability parity with the reference's manual-TP benchmark model
(reference: benchmark/torch/model/gpt_tp.py, driven by
benchmark/torch/bench_torch_tp.py): Megatron-style column/row-parallel
blocks so auto-SPMD can be compared against the best hand parallelism.

MI355X shape of the design: ONE all-reduce per block half (attention
proj, MLP proj) over the TP group — on xGMI (7 p2p links per GPU) the
per-link ring bound makes collective COUNT the cost driver, so the
conjugate identity/all-reduce pair sits exactly at the block boundary
and everything between is local. Attention heads shard across the TP
group and run through the gfx950 flash kernel locally.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from .gpt import GPTConfig


class _CopyToTP(torch.autograd.Function):
    """Megatron 'f': forward identity, backward all-reduce over TP."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, gy):
        gy = gy.contiguous()
        dist.all_reduce(gy, group=ctx.group)
        return gy, None


class _ReduceFromTP(torch.autograd.Function):
    """Megatron 'g': forward all-reduce over TP, backward identity."""

    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, gy):
        return gy, None


class ColumnParallelLinear(nn.Module):
    """Weight sharded on OUT features; output stays sharded (no gather)."""

    def __init__(self, in_f, out_f, tp, bias=True):
        super().__init__()
        assert out_f % tp == 0, (out_f, tp)
        self.weight = nn.Parameter(torch.empty(out_f // tp, in_f))
        self.bias = nn.Parameter(torch.zeros(out_f // tp)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        return F.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """Weight sharded on IN features; forward all-reduces the output."""

    def __init__(self, in_f, out_f, tp, group, bias=True):
        super().__init__()
        assert in_f % tp == 0, (in_f, tp)
        self.group = group
        self.weight = nn.Parameter(torch.empty(out_f, in_f // tp))
        self.bias = nn.Parameter(torch.zeros(out_f)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        y = F.linear(x, self.weight)           # partial sums
        y = _ReduceFromTP.apply(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


class TPSelfAttention(nn.Module):
    def __init__(self, cfg: GPTConfig, tp, group):
        super().__init__()
        assert cfg.n_head % tp == 0, (cfg.n_head, tp)
        self.n_head_local = cfg.n_head // tp
        self.n_embd = cfg.n_embd
        self.group = group
        self.c_attn = ColumnParallelLinear(cfg.n_embd, 3 * cfg.n_embd, tp,
                                           bias=cfg.bias)
        self.c_proj = RowParallelLinear(cfg.n_embd, cfg.n_embd, tp, group,
                                        bias=cfg.bias)

    def forward(self, x):
        from ..ops import attention
        B, T, C = x.shape
        x = _CopyToTP.apply(x, self.group)
        qkv = self.c_attn(x)                   # [B,T,3*C/tp]
        lc = qkv.shape[-1] // 3
        q, k, v = qkv.split(lc, dim=2)
        hd = lc // self.n_head_local
        q = q.view(B, T, self.n_head_local, hd).transpose(1, 2)
        k = k.view(B, T, self.n_head_local, hd).transpose(1, 2)
        v = v.view(B, T, self.n_head_local, hd).transpose(1, 2)
        y = attention.scaled_dot_product_attention(q, k, v, causal=True)
        y = y.transpose(1, 2).reshape(B, T, lc)
        return self.c_proj(y)


class TPMLP(nn.Module):
    def __init__(self, cfg: GPTConfig, tp, group):
        super().__init__()
        self.group = group
        self.c_fc = ColumnParallelLinear(cfg.n_embd, 4 * cfg.n_embd, tp,
                                         bias=cfg.bias)
        self.c_proj = RowParallelLinear(4 * cfg.n_embd, cfg.n_embd, tp,
                                        group, bias=cfg.bias)

    def forward(self, x):
        x = _CopyToTP.apply(x, self.group)
        return self.c_proj(F.gelu(self.c_fc(x), approximate="tanh"))


class TPBlock(nn.Module):
    def __init__(self, cfg, tp, group):
        super().__init__()
        self.ln_1 = nn.LayerNorm(cfg.n_embd, bias=cfg.bias)
        self.attn = TPSelfAttention(cfg, tp, group)
        self.ln_2 = nn.LayerNorm(cfg.n_embd, bias=cfg.bias)
        self.mlp = TPMLP(cfg, tp, group)

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x


class GPT_TP(nn.Module):
    """Manual-TP GPT. Embeddings / final LN / lm_head are replicated (the
    residual stream is replicated between block boundaries, so their
    grads agree across ranks without extra comm — Adam then keeps the
    replicas bit-identical)."""

    def __init__(self, cfg: GPTConfig, group=None):
        super().__init__()
        self.cfg = cfg
        self.group = group or dist.group.WORLD
        tp = dist.get_world_size(self.group)
        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.wpe = nn.Embedding(cfg.block_size, cfg.n_embd)
        self.h = nn.ModuleList(TPBlock(cfg, tp, self.group)
                               for _ in range(cfg.n_layer))
        self.ln_f = nn.LayerNorm(cfg.n_embd, bias=cfg.bias)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.register_buffer("pos", torch.arange(cfg.block_size),
                             persistent=False)
        nn.init.normal_(self.wte.weight, std=0.02)
        nn.init.normal_(self.wpe.weight, std=0.02)
        nn.init.normal_(self.lm_head.weight, std=0.02)

    @torch.no_grad()
    def load_from_replicated(self, gpt):
        """Shard a (replicated) models.gpt.GPT of the same config into
        this TP model: column weights split on dim 0 — q/k/v each split
        by heads —, row weights on dim 1."""
        tp = dist.get_world_size(self.group)
        r = dist.get_rank(self.group)
        E = self.cfg.n_embd
        self.wte.weight.copy_(gpt.wte.weight)
        self.wpe.weight.copy_(gpt.wpe.weight)
        self.ln_f.weight.copy_(gpt.ln_f.weight)
        if gpt.ln_f.bias is not None:
            self.ln_f.bias.copy_(gpt.ln_f.bias)
        self.lm_head.weight.copy_(gpt.lm_head.weight)
        for blk, rblk in zip(self.h, gpt.h):
            for ln, rln in ((blk.ln_1, rblk.ln_1), (blk.ln_2, rblk.ln_2)):
                ln.weight.copy_(rln.weight)
                if rln.bias is not None:
                    ln.bias.copy_(rln.bias)
            # c_attn: [3E, E] = concat(q,k,v); shard EACH of q/k/v by rows
            w = rblk.attn.c_attn.weight
            qw, kw, vw = w.split(E, dim=0)
            Ls = E // tp
            blk.attn.c_attn.weight.copy_(torch.cat(
                [t[r * Ls:(r + 1) * Ls] for t in (qw, kw, vw)], dim=0))
            if rblk.attn.c_attn.bias is not None:
                qb, kb, vb = rblk.attn.c_attn.bias.split(E, dim=0)
                blk.attn.c_attn.bias.copy_(torch.cat(
                    [t[r * Ls:(r + 1) * Ls] for t in (qb, kb, vb)], dim=0))
            blk.attn.c_proj.weight.copy_(
                rblk.attn.c_proj.weight[:, r * Ls:(r + 1) * Ls])
            if rblk.attn.c_proj.bias is not None:
                blk.attn.c_proj.bias.copy_(rblk.attn.c_proj.bias)
            H = 4 * E // tp
            blk.mlp.c_fc.weight.copy_(
                rblk.mlp.c_fc.weight[r * H:(r + 1) * H])
            if rblk.mlp.c_fc.bias is not None:
                blk.mlp.c_fc.bias.copy_(
                    rblk.mlp.c_fc.bias[r * H:(r + 1) * H])
            blk.mlp.c_proj.weight.copy_(
                rblk.mlp.c_proj.weight[:, r * H:(r + 1) * H])
            if rblk.mlp.c_proj.bias is not None:
                blk.mlp.c_proj.bias.copy_(rblk.mlp.c_proj.bias)

    def forward(self, idx):
        B, T = idx.shape
        x = self.wte(idx) + self.wpe(self.pos[:T])
        if torch.is_autocast_enabled(x.device.type):
            x = x.to(torch.bfloat16)
        for blk in self.h:
            x = blk(x)
        x = self.ln_f(x)
        return self.lm_head(x)

    def loss(self, idx, targets):
        logits = self(idx)
        return F.cross_entropy(logits.view(-1, logits.size(-1)),
                               targets.reshape(-1))
