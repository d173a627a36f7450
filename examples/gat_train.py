#!/usr/bin/env python3
"""Graph attention network node classification through auto-SPMD
(reference: examples/torch/gnn/{gat,train}.py; synthetic random graph —
no dataset downloads in this environment).

    torchrun --nproc_per_node N examples/gat_train.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn.functional as F

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
from easydist_amd.models.gat import GAT


def train_step(model, opt, x, adj, y):
    loss = F.cross_entropy(model(x, adj), y)
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
    os.environ.setdefault("MASTER_PORT", "29563")
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    easydist_setup(backend="torch", device="cuda" if use_cuda else "cpu")
    set_device_mesh(list(range(world)), ["spmd0"])

    torch.manual_seed(0)
    n, feat = (512, 256) if not use_cuda else (4096, 4096)
    model = GAT(in_dim=feat, hidden=128).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=use_cuda)
    compiled = easydist_compile(train_step)

    g = torch.Generator().manual_seed(7)
    x = torch.randn(n, feat, generator=g).to(device)
    adj = (torch.rand(n, n, generator=g) < 0.02).float().to(device)
    y = torch.randint(0, 64, (n,), generator=g).to(device)
    for step in range(10):
        loss = compiled(model, opt, x, adj, y)
        if rank == 0 and step % 2 == 0:
            print(f"step {step:3d}  loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
