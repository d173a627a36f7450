#!/usr/bin/env python3
"""Mixtral-shape MoE with expert parallelism over xGMI all-to-all.

    torchrun --nproc_per_node N examples/moe_ep_train.py
(n_experts must be divisible by N.)
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from easydist_amd.models.moe import MoEConfig, MoEGPT, moe_train_step


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

    cfg = MoEConfig(vocab_size=4096, n_layer=2, n_head=8, n_embd=512,
                    block_size=128, n_experts=8, top_k=2, ffn_hidden=1024)
    torch.manual_seed(0)
    model = MoEGPT(cfg, ep_group=dist.group.WORLD if world > 1 else None)
    model = model.to(device)
    opt = torch.optim.Adam(model.parameters(), lr=3e-4, fused=use_cuda)

    g = torch.Generator().manual_seed(rank)   # each rank: own tokens
    for step in range(10):
        idx = torch.randint(0, cfg.vocab_size, (8, 128), generator=g).to(device)
        tg = torch.randint(0, cfg.vocab_size, (8, 128), generator=g).to(device)
        loss = moe_train_step(model, opt, idx, tg)
        if rank == 0 and step % 2 == 0:
            print(f"step {step:2d}  loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
