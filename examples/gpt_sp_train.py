#!/usr/bin/env python3
"""Sequence-parallel GPT training demo: each rank holds a sequence
shard; attention runs as exact ring attention over xGMI neighbors.

Launch: torchrun --nproc_per_node N --master-addr 127.0.0.1 \
            examples/gpt_sp_train.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from easydist_amd.models.gpt import GPT, GPTConfig


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29552")
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(rank % torch.cuda.device_count())
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    device = "cuda" if use_cuda else "cpu"

    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=50304, n_layer=4, n_head=8, n_embd=512,
                    block_size=2048)
    model = GPT(cfg).to(device)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model.enable_sequence_parallel(dist.group.WORLD)
    opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=use_cuda)

    S_local = cfg.block_size // world
    for step in range(10):
        # synthetic batch; every rank slices ITS sequence shard
        g = torch.Generator().manual_seed(100 + step)
        idx = torch.randint(0, cfg.vocab_size, (2, cfg.block_size),
                            generator=g).to(device)
        tg = torch.randint(0, cfg.vocab_size, (2, cfg.block_size),
                           generator=g).to(device)
        sl = slice(rank * S_local, (rank + 1) * S_local)
        logits = model(idx[:, sl])
        lsum = torch.nn.functional.cross_entropy(
            logits.reshape(-1, cfg.vocab_size), tg[:, sl].reshape(-1),
            reduction="sum")
        loss = lsum / tg.numel()
        loss.backward()
        # replicated params: grads are partial over the sequence split
        for p in model.parameters():
            if p.grad is not None:
                dist.all_reduce(p.grad)
        opt.step()
        opt.zero_grad(True)
        total = loss.detach().clone()
        dist.all_reduce(total)
        if rank == 0:
            print(f"step {step} loss {float(total):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
