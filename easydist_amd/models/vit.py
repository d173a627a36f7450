"""Vision Transformer for the ViT-Large BASELINE config.

BASELINE.json: "ViT-Large 4k-batch auto-SPMD (mixed DP+TP,
reduce-scatter reshard) on 8x MI355X". Attention reuses the gfx950 flash
kernel (non-causal); no torchvision dependency (not in the image).
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class ViTConfig:
    image_size: int = 224
    patch_size: int = 16
    n_layer: int = 24
    n_head: int = 16
    n_embd: int = 1024
    mlp_ratio: int = 4
    n_classes: int = 1000
    in_chans: int = 3


VIT_LARGE = ViTConfig()
VIT_BASE = ViTConfig(n_layer=12, n_head=12, n_embd=768)
VIT_TINY = ViTConfig(image_size=32, patch_size=8, n_layer=2, n_head=2,
                     n_embd=64, n_classes=10)


class ViTAttention(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.n_head = cfg.n_head
        self.qkv = nn.Linear(cfg.n_embd, 3 * cfg.n_embd)
        self.proj = nn.Linear(cfg.n_embd, cfg.n_embd)

    def forward(self, x):
        from ..ops import attention
        B, T, C = x.shape
        q, k, v = self.qkv(x).split(C, dim=2)
        hd = C // self.n_head
        q = q.view(B, T, self.n_head, hd).transpose(1, 2)
        k = k.view(B, T, self.n_head, hd).transpose(1, 2)
        v = v.view(B, T, self.n_head, hd).transpose(1, 2)
        y = attention.scaled_dot_product_attention(q, k, v, causal=False)
        return self.proj(y.transpose(1, 2).reshape(B, T, C))


class ViTBlock(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.ln_1 = nn.LayerNorm(cfg.n_embd)
        self.attn = ViTAttention(cfg)
        self.ln_2 = nn.LayerNorm(cfg.n_embd)
        h = cfg.n_embd * cfg.mlp_ratio
        self.mlp = nn.Sequential(nn.Linear(cfg.n_embd, h), nn.GELU(),
                                 nn.Linear(h, cfg.n_embd))

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x


class ViT(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.cfg = cfg
        n_patches = (cfg.image_size // cfg.patch_size) ** 2
        self.patch_embed = nn.Conv2d(cfg.in_chans, cfg.n_embd,
                                     cfg.patch_size, stride=cfg.patch_size)
        self.cls = nn.Parameter(torch.zeros(1, 1, cfg.n_embd))
        self.pos = nn.Parameter(torch.zeros(1, n_patches + 1, cfg.n_embd))
        self.blocks = nn.ModuleList(ViTBlock(cfg)
                                    for _ in range(cfg.n_layer))
        self.ln_f = nn.LayerNorm(cfg.n_embd)
        self.head = nn.Linear(cfg.n_embd, cfg.n_classes)
        nn.init.trunc_normal_(self.pos, std=0.02)
        nn.init.trunc_normal_(self.cls, std=0.02)

    def forward(self, x):
        B = x.shape[0]
        x = self.patch_embed(x).flatten(2).transpose(1, 2)   # [B, N, C]
        x = torch.cat([self.cls.expand(B, -1, -1).to(x.dtype), x],
                      dim=1) + self.pos.to(x.dtype)
        for blk in self.blocks:
            x = blk(x)
        return self.head(self.ln_f(x)[:, 0])


def vit_train_step(model, opt, x, y):
    dev = "cuda" if x.is_cuda else "cpu"
    with torch.autocast(device_type=dev, dtype=torch.bfloat16,
                        enabled=x.is_cuda):
        logits = model(x)
    loss = F.cross_entropy(logits.float(), y)
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss
