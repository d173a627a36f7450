#!/usr/bin/env python3
"""Smallest end-to-end example: an MLP train step through the full
auto-SPMD pipeline (reference: examples/torch/simple_model.py).

    torchrun --nproc_per_node N examples/simple_model.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn as nn

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh


def train_step(model, opt, x, y):
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss


def main():
    use_cuda = torch.cuda.is_available()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    device = "cuda" if use_cuda else "cpu"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29565")
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    easydist_setup(backend="torch", device=device)
    set_device_mesh(list(range(world)), ["spmd0"])

    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(128, 256), nn.ReLU(),
                          nn.Linear(256, 64)).to(device)
    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    compiled = easydist_compile(train_step)

    g = torch.Generator().manual_seed(7)
    for step in range(10):
        x = torch.randn(64, 128, generator=g).to(device)
        y = torch.randn(64, 64, generator=g).to(device)
        loss = compiled(model, opt, x, y)
        if rank == 0 and step % 2 == 0:
            print(f"step {step:3d}  loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
