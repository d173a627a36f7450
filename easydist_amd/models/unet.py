"""Toy DDPM-style UNet for the diffusion example.

Capability parity target: the reference ships a stable-diffusion demo
(examples/torch/stable_diffusion.py) driving a pretrained UNet through
its pipeline. There is no network (or weights) in this environment, so
the MI355X port demonstrates the same capability — a skip-connected
conv/transpose-conv UNet with timestep conditioning through the full
trace→solve→shard pipeline — on a random-init model and synthetic
latents. All ops (convolution fwd/bwd incl. transposed, silu, cat,
linear) are covered by the analytic preset layer.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


def timestep_embedding(t: torch.Tensor, dim: int) -> torch.Tensor:
    """Sinusoidal embedding, [B] -> [B, dim]."""
    half = dim // 2
    freqs = torch.exp(
        -math.log(10000.0) * torch.arange(half, dtype=torch.float32,
                                          device=t.device) / half)
    ang = t.float()[:, None] * freqs[None, :]
    return torch.cat([torch.cos(ang), torch.sin(ang)], dim=-1)


class Block(nn.Module):
    def __init__(self, cin, cout, tdim):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 3, padding=1)
        self.conv2 = nn.Conv2d(cout, cout, 3, padding=1)
        self.temb = nn.Linear(tdim, cout)
        self.skip = (nn.Conv2d(cin, cout, 1) if cin != cout
                     else nn.Identity())

    def forward(self, x, temb):
        h = F.silu(self.conv1(x))
        h = h + self.temb(temb)[:, :, None, None]
        h = F.silu(self.conv2(h))
        return h + self.skip(x)


class ToyUNet(nn.Module):
    """3-level UNet: stride-2 conv down, transpose-conv up, skip cats."""

    def __init__(self, cin: int = 4, base: int = 64, tdim: int = 128):
        super().__init__()
        self.tdim = tdim
        self.tmlp = nn.Sequential(nn.Linear(tdim, tdim), nn.SiLU(),
                                  nn.Linear(tdim, tdim))
        self.inp = nn.Conv2d(cin, base, 3, padding=1)
        self.d1 = Block(base, base, tdim)
        self.down1 = nn.Conv2d(base, base * 2, 3, stride=2, padding=1)
        self.d2 = Block(base * 2, base * 2, tdim)
        self.down2 = nn.Conv2d(base * 2, base * 4, 3, stride=2, padding=1)
        self.mid = Block(base * 4, base * 4, tdim)
        self.up2 = nn.ConvTranspose2d(base * 4, base * 2, 4, stride=2,
                                      padding=1)
        self.u2 = Block(base * 4, base * 2, tdim)
        self.up1 = nn.ConvTranspose2d(base * 2, base, 4, stride=2, padding=1)
        self.u1 = Block(base * 2, base, tdim)
        self.out = nn.Conv2d(base, cin, 3, padding=1)

    def forward(self, x, t):
        temb = self.tmlp(timestep_embedding(t, self.tdim))
        h0 = self.inp(x)
        h1 = self.d1(h0, temb)
        h2 = self.d2(self.down1(h1), temb)
        hm = self.mid(self.down2(h2), temb)
        u2 = self.u2(torch.cat([self.up2(hm), h2], dim=1), temb)
        u1 = self.u1(torch.cat([self.up1(u2), h1], dim=1), temb)
        return self.out(u1)


def ddpm_train_step(model, opt, x0, t, noise, abar):
    """One denoising-diffusion step: predict the noise added at t.

    abar: [B] cumulative alpha-bar gathered for each sample's t."""
    a = abar[:, None, None, None]
    xt = a.sqrt() * x0 + (1 - a).sqrt() * noise
    pred = model(xt, t)
    loss = ((pred - noise) ** 2).mean()
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss
