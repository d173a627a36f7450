#!/usr/bin/env python3
"""DDP parallel mode (manual graph transform, not the solver): the same
train step compiled with parallel_mode="ddp" (reference:
examples/torch/simple_ddp.py).

    torchrun --nproc_per_node N examples/simple_ddp.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn as nn

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh


def train_step(model, opt, x, y):
    loss = torch.nn.functional.cross_entropy(model(x), y)
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
    os.environ.setdefault("MASTER_PORT", "29569")
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    easydist_setup(backend="torch", device=device)
    set_device_mesh(list(range(world)), ["spmd0"])

    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(784, 512), nn.ReLU(),
                          nn.Linear(512, 10)).to(device)
    # ddp mode consumes the UN-decomposed fused-optimizer graph
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    compiled = easydist_compile(train_step, parallel_mode="ddp")

    g = torch.Generator().manual_seed(7 + rank)   # per-rank data in ddp
    for step in range(10):
        x = torch.randn(32, 784, generator=g).to(device)
        y = torch.randint(0, 10, (32,), generator=g).to(device)
        loss = compiled(model, opt, x, y)
        if rank == 0 and step % 2 == 0:
            print(f"step {step:3d}  loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
