#!/usr/bin/env python3
"""Minimal auto-SPMD GPT training (reference: examples/torch/gpt_train.py).

    torchrun --nproc_per_node N examples/gpt_train.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
from easydist_amd.models import gpt as gptm


def main():
    use_cuda = torch.cuda.is_available()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local = int(os.environ.get("LOCAL_RANK", 0))
    device = f"cuda:{local}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29560")
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    easydist_setup(backend="torch", device="cuda" if use_cuda else "cpu")
    set_device_mesh(list(range(world)), ["spmd0"])

    cfg = gptm.GPTConfig(vocab_size=4096, n_layer=4, n_head=8, n_embd=512,
                         block_size=256)
    torch.manual_seed(0)
    model = gptm.GPT(cfg).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=3e-4, fused=use_cuda)
    compiled = easydist_compile(gptm.gpt_train_step, cuda_graph=use_cuda)

    g = torch.Generator().manual_seed(7)
    for step in range(20):
        idx = torch.randint(0, cfg.vocab_size, (16, 256), generator=g).to(device)
        tg = torch.randint(0, cfg.vocab_size, (16, 256), generator=g).to(device)
        loss = compiled(model, opt, idx, tg)
        if rank == 0 and step % 5 == 0:
            print(f"step {step:3d}  loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
