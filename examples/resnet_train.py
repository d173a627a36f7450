#!/usr/bin/env python3
"""WideResNet-50 image classification through auto-SPMD (reference:
examples/torch/resnet_train.py; synthetic data — no dataset downloads
in this environment).

    torchrun --nproc_per_node N examples/resnet_train.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn.functional as F

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
from easydist_amd.models.resnet import wresnet50


def train_step(model, opt, x, y):
    loss = F.cross_entropy(model(x), y)
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss


def main():
    use_cuda = torch.cuda.is_available()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local = int(os.environ.get("LOCAL_RANK", 0))
    device = f"cuda:{local}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29562")
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    easydist_setup(backend="torch", device="cuda" if use_cuda else "cpu")
    set_device_mesh(list(range(world)), ["spmd0"])

    torch.manual_seed(0)
    model = wresnet50().to(device)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    compiled = easydist_compile(train_step)

    B, res = (4, 64) if not use_cuda else (64, 224)
    g = torch.Generator().manual_seed(7)
    for step in range(10):
        x = torch.randn(B, 3, res, res, generator=g).to(device)
        y = torch.randint(0, 1000, (B,), generator=g).to(device)
        loss = compiled(model, opt, x, y)
        if rank == 0 and step % 2 == 0:
            print(f"step {step:3d}  loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
