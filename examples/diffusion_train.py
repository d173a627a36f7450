#!/usr/bin/env python3
"""DDPM-style diffusion training through auto-SPMD (capability parity
with reference examples/torch/stable_diffusion.py — no network here, so
a random-init toy UNet on synthetic latents exercises the same path:
conv/transpose-conv UNet + timestep conditioning, traced, solved and
sharded like any other model).

    torchrun --nproc_per_node N examples/diffusion_train.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
from easydist_amd.models.unet import ToyUNet, ddpm_train_step


def main():
    use_cuda = torch.cuda.is_available()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local = int(os.environ.get("LOCAL_RANK", 0))
    device = f"cuda:{local}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29561")
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    easydist_setup(backend="torch", device="cuda" if use_cuda else "cpu")
    set_device_mesh(list(range(world)), ["spmd0"])

    torch.manual_seed(0)
    model = ToyUNet(cin=4, base=32 if not use_cuda else 64).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=use_cuda)
    compiled = easydist_compile(ddpm_train_step)

    T = 1000
    betas = torch.linspace(1e-4, 0.02, T)
    abar_all = torch.cumprod(1 - betas, 0).to(device)
    B, res = (8, 32) if not use_cuda else (32, 64)
    g = torch.Generator().manual_seed(7)
    for step in range(10):
        x0 = torch.randn(B, 4, res, res, generator=g).to(device)
        t = torch.randint(0, T, (B,), generator=g).to(device)
        noise = torch.randn(B, 4, res, res, generator=g).to(device)
        loss = compiled(model, opt, x0, t, noise, abar_all[t])
        if rank == 0 and step % 2 == 0:
            print(f"step {step:3d}  loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
