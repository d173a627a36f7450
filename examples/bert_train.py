#!/usr/bin/env python3
"""Train a HuggingFace BERT (masked-LM) through auto-SPMD.

Mirrors reference examples/torch/bert_train.py: an UNMODIFIED
transformers model + torch optimizer handed to easydist_compile; the
whole step (fwd+loss+bwd+adam) is traced, sharded over the mesh, and
run with RCCL collectives.

Launch: torchrun --nproc_per_node N --master-addr 127.0.0.1 \
            examples/bert_train.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
from transformers import BertConfig, BertForMaskedLM

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh


def train_step(model, opt, input_ids, labels):
    loss = model(input_ids=input_ids, labels=labels).loss
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29551")
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(rank % torch.cuda.device_count())
    dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank,
                            world_size=world)
    device = "cuda" if use_cuda else "cpu"
    easydist_setup(backend="torch", device=device)
    set_device_mesh(list(range(world)), ["spmd0"])

    torch.manual_seed(42)
    cfg = BertConfig()          # bert-base geometry, random init
    model = BertForMaskedLM(cfg).to(device)
    model.train()
    opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=use_cuda)
    compiled = easydist_compile(train_step, cuda_graph=use_cuda)

    for step in range(10):
        ids = torch.randint(0, cfg.vocab_size, (8, 128), device=device)
        labels = ids.clone()
        loss = compiled(model, opt, ids, labels)
        if rank == 0:
            print(f"step {step} loss {float(loss):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
