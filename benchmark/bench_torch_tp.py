#!/usr/bin/env python3
"""Manual-TP GPT vs auto-SPMD comparison harness.

reference: benchmark/torch/bench_torch_tp.py + model/gpt_tp.py — the
hand-written Megatron-style tensor-parallel GPT is the strongest manual
baseline the auto compiler is measured against.

Launch:  torchrun --nproc_per_node N --master-addr 127.0.0.1 \
             benchmark/bench_torch_tp.py --mode tp|auto [--steps 10]
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
    ap.add_argument("--mode", default="tp", choices=["tp", "auto"])
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--hidden", type=int, default=0,
                    help="override n_embd (CPU smoke)")
    ap.add_argument("--seq", type=int, default=1024)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29541")
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(rank % torch.cuda.device_count())
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    device = f"cuda:{rank % torch.cuda.device_count()}" if use_cuda \
        else "cpu"

    from easydist_amd.models.gpt import GPT, GPT_BENCH_1L
    torch.manual_seed(42)
    cfg = GPT_BENCH_1L
    if args.hidden:
        from easydist_amd.models.gpt import GPTConfig
        cfg = GPTConfig(vocab_size=1024, n_layer=1,
                        n_head=max(4, args.hidden // 64),
                        n_embd=args.hidden, block_size=args.seq)
    x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                      device=device)
    y = torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                      device=device)

    if args.mode == "tp":
        from easydist_amd.models.gpt_tp import GPT_TP
        model = GPT_TP(cfg).to(device)
        opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=use_cuda)

        def run():
            from easydist_amd.ops import ce
            with torch.autocast(device_type="cuda" if use_cuda else "cpu",
                                dtype=torch.bfloat16):
                logits = model(x)
            loss = ce.cross_entropy(logits.view(-1, logits.size(-1)),
                                    y.reshape(-1))
            loss.backward()
            opt.step()
            opt.zero_grad(True)
            return loss
    else:
        from easydist_amd import (easydist_compile, easydist_setup,
                                  set_device_mesh)
        from easydist_amd.models.gpt import gpt_train_step
        easydist_setup(backend="torch", device="cuda" if use_cuda else "cpu")
        set_device_mesh(list(range(world)), ["spmd0"])
        model = GPT(cfg).to(device)
        opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=use_cuda)
        compiled = easydist_compile(gpt_train_step, cuda_graph=use_cuda)

        def run():
            return compiled(model, opt, x, y)

    for _ in range(args.warmup):
        run()
    if use_cuda:
        torch.cuda.synchronize()
        torch.cuda.reset_peak_memory_stats()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run()
    if use_cuda:
        torch.cuda.synchronize()
    dist.barrier()
    dt = (time.perf_counter() - t0) / args.steps
    peak = (torch.cuda.max_memory_allocated() / 2**30) if use_cuda else 0.0
    t = torch.tensor([dt], device=device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    if rank == 0:
        print(json.dumps({
            "bench": "gpt_tp_compare", "mode": args.mode,
            "world_size": world, "ms_per_step": t.item() * 1e3,
            "samples_per_sec": args.batch * world / t.item(),
            "peak_mem_gb": round(peak, 2)}))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
