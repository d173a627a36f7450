"""GPT-2 family for training benchmarks (synthetic data, random init).

Mirrors the reference's benchmark GPT config surface
(reference: benchmark/bench_case.py:5-13, benchmark/torch/model/gpt.py) with
an MI355X-friendly formulation: attention goes through
``easydist_amd.ops.attention`` (hand-written HIP flash kernel on gfx950,
aten SDPA fallback on CPU), norms through ``easydist_amd.ops.layer_norm``.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class GPTConfig:
    vocab_size: int = 50304          # padded to a multiple of 128
    n_layer: int = 12
    n_head: int = 12
    n_embd: int = 768
    block_size: int = 1024
    dropout: float = 0.0             # benchmarks run dropout-free
    bias: bool = True


GPT2_SMALL = GPTConfig()
GPT2_MEDIUM = GPTConfig(n_layer=24, n_head=16, n_embd=1024)
GPT2_LARGE = GPTConfig(n_layer=36, n_head=20, n_embd=1280)
GPT2_XL = GPTConfig(n_layer=48, n_head=25, n_embd=1600)
GPT2_1_3B = GPTConfig(n_layer=24, n_head=32, n_embd=2048, block_size=1024)
# the reference's single-layer benchmark shape (bench_case.py:5-13)
GPT_BENCH_1L = GPTConfig(n_layer=1, n_head=48, n_embd=12288, block_size=1024)


class CausalSelfAttention(nn.Module):
    """Sequence-parallel ready: with ``sp_group`` set (see
    GPT.enable_sequence_parallel) each rank holds a sequence shard and
    attention runs as exact ring attention over the xGMI neighbors
    (ops/ring_attention.py); otherwise the gfx950 flash kernel handles
    the full sequence locally."""

    def __init__(self, cfg: GPTConfig):
        super().__init__()
        assert cfg.n_embd % cfg.n_head == 0
        self.n_head = cfg.n_head
        self.n_embd = cfg.n_embd
        self.sp_group = None
        self.c_attn = nn.Linear(cfg.n_embd, 3 * cfg.n_embd, bias=cfg.bias)
        self.c_proj = nn.Linear(cfg.n_embd, cfg.n_embd, bias=cfg.bias)

    def forward(self, x):
        from ..ops import attention, ring_attention
        B, T, C = x.shape
        qkv = self.c_attn(x)
        q, k, v = qkv.split(self.n_embd, dim=2)
        hd = C // self.n_head
        q = q.view(B, T, self.n_head, hd).transpose(1, 2)
        k = k.view(B, T, self.n_head, hd).transpose(1, 2)
        v = v.view(B, T, self.n_head, hd).transpose(1, 2)
        if self.sp_group is not None:
            y = ring_attention.ring_attention(q, k, v, group=self.sp_group,
                                              causal=True)
        else:
            y = attention.scaled_dot_product_attention(q, k, v, causal=True)
        y = y.transpose(1, 2).reshape(B, T, C)
        return self.c_proj(y)


class MLPBlock(nn.Module):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.c_fc = nn.Linear(cfg.n_embd, 4 * cfg.n_embd, bias=cfg.bias)
        self.c_proj = nn.Linear(4 * cfg.n_embd, cfg.n_embd, bias=cfg.bias)

    def forward(self, x):
        return self.c_proj(F.gelu(self.c_fc(x), approximate="tanh"))


class Block(nn.Module):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.ln_1 = nn.LayerNorm(cfg.n_embd, bias=cfg.bias)
        self.attn = CausalSelfAttention(cfg)
        self.ln_2 = nn.LayerNorm(cfg.n_embd, bias=cfg.bias)
        self.mlp = MLPBlock(cfg)

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x


class GPT(nn.Module):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.wpe = nn.Embedding(cfg.block_size, cfg.n_embd)
        self.h = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layer))
        self.ln_f = nn.LayerNorm(cfg.n_embd, bias=cfg.bias)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.register_buffer("pos", torch.arange(cfg.block_size),
                             persistent=False)
        self.apply(self._init)
        for n, p in self.named_parameters():
            if n.endswith("c_proj.weight"):
                nn.init.normal_(p, std=0.02 / math.sqrt(2 * cfg.n_layer))

    @staticmethod
    def _init(m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def enable_sequence_parallel(self, group):
        """Sequence-parallel mode: every rank feeds its OWN sequence
        shard (token slice) of the batch; attention rings over the
        group; positions offset by the rank's slice start."""
        import torch.distributed as dist
        self._sp_group = group
        self._sp_rank = dist.get_rank(group)
        for blk in self.h:
            blk.attn.sp_group = group
        return self

    def forward(self, idx):
        B, T = idx.shape
        off = getattr(self, "_sp_rank", 0) * T \
            if getattr(self, "_sp_group", None) is not None else 0
        x = self.wte(idx) + self.wpe(self.pos[off:off + T])
        if torch.is_autocast_enabled(x.device.type):
            # embeddings are not on autocast's cast list: without this the
            # whole residual stream runs fp32 (fp32 norms/adds + a cast
            # pair around every matmul)
            x = x.to(torch.bfloat16)
        for blk in self.h:
            x = blk(x)
        x = self.ln_f(x)
        return self.lm_head(x)

    def loss(self, idx, targets):
        logits = self(idx)
        return F.cross_entropy(logits.view(-1, logits.size(-1)),
                               targets.reshape(-1))


def gpt_train_step(model, opt, idx, targets, autocast_device=None):
    """The benchmarked train step: fwd+loss+bwd+adam, bf16 autocast.

    The loss goes through the fused HIP cross-entropy (ops/ce.py): one
    kernel computes logsumexp+NLL straight from the bf16 logits instead
    of materializing a [tokens, vocab] fp32 log-softmax (1.6 GB at GPT-2
    shapes) — lse/softmax reductions still accumulate in fp32 inside the
    kernel."""
    from ..ops import ce
    dev = autocast_device or ("cuda" if idx.is_cuda else "cpu")
    with torch.autocast(device_type=dev, dtype=torch.bfloat16):
        logits = model(idx)
    loss = ce.cross_entropy(logits.view(-1, logits.size(-1)),
                            targets.reshape(-1))
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss
