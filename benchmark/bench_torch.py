#!/usr/bin/env python3
"""Comparison harness: easydist_amd auto-SPMD vs torch DDP vs FSDP.

reference: benchmark/torch/bench_torch.py (train-step wall time + peak
GPU memory for GPT / WideResNet / GAT cases, bench_case.py shapes).

Launch:  torchrun --nproc_per_node N benchmark/bench_torch.py \
             --model gpt --mode auto|ddp|fsdp|zero2|zero3 [--steps 10]
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


def build_case(name: str, device):
    torch.manual_seed(42)
    if name == "gpt":
        from easydist_amd.models.gpt import GPT, GPT_BENCH_1L
        model = GPT(GPT_BENCH_1L).to(device)
        x = torch.randint(0, GPT_BENCH_1L.vocab_size, (4, 1024),
                          device=device)
        y = torch.randint(0, GPT_BENCH_1L.vocab_size, (4, 1024),
                          device=device)

        def step(model, opt, x, y):
            from easydist_amd.ops import ce
            logits = model(x)
            loss = ce.cross_entropy(logits.view(-1, logits.size(-1)),
                                    y.reshape(-1))
            loss.backward()
            opt.step()
            opt.zero_grad(True)
            return loss
        return model, step, (x, y)
    if name == "wresnet":
        from easydist_amd.models.resnet import wresnet50
        model = wresnet50().to(device)
        x = torch.randn(128, 3, 224, 224, device=device)
        y = torch.randint(0, 1000, (128,), device=device)

        def step(model, opt, x, y):
            loss = torch.nn.functional.cross_entropy(model(x), y)
            loss.backward()
            opt.step()
            opt.zero_grad(True)
            return loss
        return model, step, (x, y)
    if name == "gat":
        from easydist_amd.models.gat import GAT
        model = GAT(in_dim=12288, hidden=512).to(device)
        x = torch.randn(4096, 12288, device=device)
        adj = (torch.rand(4096, 4096, device=device) < 0.01).float()
        y = torch.randint(0, 64, (4096,), device=device)

        def step(model, opt, x, y, adj):
            loss = torch.nn.functional.cross_entropy(model(x, adj), y)
            loss.backward()
            opt.step()
            opt.zero_grad(True)
            return loss
        return model, step, (x, y, adj)
    raise ValueError(name)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt",
                    choices=["gpt", "wresnet", "gat"])
    ap.add_argument("--mode", default="auto",
                    choices=["auto", "ddp", "fsdp", "zero2", "zero3",
                             "vanilla"])
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    args = ap.parse_args()

    use_cuda = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", 0))
    local = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    device = f"cuda:{local}" if use_cuda else "cpu"
    if args.mode in ("auto", "zero2", "zero3"):
        # BEFORE any CUDA context init: easydist_setup swaps in the
        # profiling allocator (EASYDIST_MEM_OPT), which torch rejects
        # once the default allocator has served an allocation
        from easydist_amd import easydist_setup
        easydist_setup(backend="torch", device="cuda" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(local)
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29521")
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world)

    model, step, inputs = build_case(args.model, device)

    if args.mode in ("auto", "zero2", "zero3"):
        from easydist_amd import easydist_compile, set_device_mesh
        set_device_mesh(list(range(world)), ["spmd0"])
        opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=use_cuda)
        mode = args.mode if args.mode != "auto" else "auto"
        compiled = easydist_compile(step, parallel_mode=mode,
                                    cuda_graph=False)
        run = lambda: compiled(model, opt, *inputs)  # noqa: E731
    elif args.mode == "ddp":
        m = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local] if use_cuda else None)
        opt = torch.optim.Adam(m.parameters(), lr=1e-4, fused=use_cuda)
        run = lambda: step(m, opt, *inputs)          # noqa: E731
    elif args.mode == "fsdp":
        from torch.distributed.fsdp import FullyShardedDataParallel
        m = FullyShardedDataParallel(model)
        opt = torch.optim.Adam(m.parameters(), lr=1e-4)
        run = lambda: step(m, opt, *inputs)          # noqa: E731
    else:
        opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=use_cuda)
        run = lambda: step(model, opt, *inputs)      # noqa: E731

    for _ in range(args.warmup):
        run()
    if use_cuda:
        torch.cuda.synchronize()
        try:
            torch.cuda.reset_peak_memory_stats()
        except RuntimeError:
            pass          # pluggable allocator: no torch-side peak stats
    t0 = time.time()
    for _ in range(args.steps):
        run()
    if use_cuda:
        torch.cuda.synchronize()
    dt = (time.time() - t0) / args.steps
    peak = 0.0
    if use_cuda:
        try:
            peak = torch.cuda.max_memory_allocated() / 2**30
        except RuntimeError:
            # static-plan playback: the packed arena IS the footprint
            from easydist_amd.memory import _mem_alloc
            peak = _mem_alloc.arena_size() / 2**30
    if rank == 0:
        print(json.dumps({"model": args.model, "mode": args.mode,
                          "ms_per_step": dt * 1000,
                          "peak_mem_gb": round(peak, 2),
                          "world": world}))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
