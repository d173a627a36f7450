#!/usr/bin/env python3
"""Pipeline-parallel throughput sweep: GPipe vs DAPPLE vs vanilla.

reference: benchmark/torch/pp/{gpt,resnet101}/speed/batch.sh — microbatch
count sweep on GPT; launch with torchrun --nproc_per_node <stages>.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--schedule", default="dapple",
                    choices=["gpipe", "dapple", "vanilla"])
    ap.add_argument("--nchunks", type=int, default=8)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--layers", type=int, default=12)
    args = ap.parse_args()

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.gpt import GPT, GPTConfig
    from easydist_amd.ops import ce

    use_cuda = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", 0))
    local = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    device = f"cuda:{local}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local)
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29522")
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world)
    easydist_setup(backend="torch", device="cuda" if use_cuda else "cpu")
    set_device_mesh(list(range(world)), ["spmd0"])

    cfg = GPTConfig(n_layer=args.layers, block_size=512)
    torch.manual_seed(0)
    model = GPT(cfg).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-4)

    def train_step(model, opt, idx, tg):
        logits = model(idx)
        loss = ce.cross_entropy(logits.view(-1, logits.size(-1)),
                                tg.reshape(-1))
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    idx = torch.randint(0, cfg.vocab_size, (args.batch, 512), device=device)
    tg = torch.randint(0, cfg.vocab_size, (args.batch, 512), device=device)

    if args.schedule == "vanilla":
        run = lambda: train_step(model, opt, idx, tg)    # noqa: E731
    else:
        compiled = easydist_compile(train_step, parallel_mode="pp",
                                    cuda_graph=False, nstages=world,
                                    nchunks=args.nchunks,
                                    schedule=args.schedule)
        run = lambda: compiled(model, opt, idx, tg)      # noqa: E731

    for _ in range(3):
        run()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        run()
    if use_cuda:
        torch.cuda.synchronize()
    dt = (time.time() - t0) / args.steps
    if rank == 0:
        print(json.dumps({"schedule": args.schedule,
                          "nchunks": args.nchunks,
                          "samples_per_sec": args.batch / dt,
                          "ms_per_step": dt * 1000, "stages": world}))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
