#!/usr/bin/env python3
"""Compile a BARE function — no nn.Module, no optimizer (reference:
examples/torch/simple_function.py): the pipeline shards any traceable
tensor program, not just training steps.

    torchrun --nproc_per_node N examples/simple_function.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh


def fn(a, b, w):
    h = torch.mm(a, w).relu()
    s = torch.mm(h, b)
    return s.sum(dim=0)


def main():
    use_cuda = torch.cuda.is_available()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    device = "cuda" if use_cuda else "cpu"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29564")
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    easydist_setup(backend="torch", device=device)
    set_device_mesh(list(range(world)), ["spmd0"])

    torch.manual_seed(0)
    a = torch.randn(512, 256, device=device)
    b = torch.randn(128, 64, device=device)
    w = torch.randn(256, 128, device=device)
    compiled = easydist_compile(fn)
    out = compiled(a, b, w)
    ref = fn(a, b, w)
    if rank == 0:
        print("max |compiled - eager| =",
              float((out - ref).abs().max()))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
