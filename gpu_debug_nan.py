"""Bisect hipGraph-replay corruption: loss + param checksum per step."""
import os
import sys

import torch
import torch.distributed as dist

from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
from easydist_amd.models import gpt as gptm


def run(use_graph: bool):
    torch.manual_seed(1234)
    cfg = gptm.GPTConfig(vocab_size=512, n_layer=2, n_head=4, n_embd=256,
                         block_size=128)
    model = gptm.GPT(cfg).cuda()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    compiled = easydist_compile(gptm.gpt_train_step,
                                cuda_graph=use_graph)
    g = torch.Generator(device="cpu").manual_seed(7)
    idx = torch.randint(0, 512, (8, 128), generator=g).cuda()
    tg = torch.randint(0, 512, (8, 128), generator=g).cuda()
    out = []
    for step in range(6):
        loss = compiled(model, opt, idx, tg)
        rt = list(compiled.compiled.values())[0]
        csum = sum(float(t.float().sum()) for t in rt.state.values()
                   if t.is_floating_point())
        out.append((float(loss), csum))
    return out


if __name__ == "__main__":
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29598")
    dist.init_process_group(backend="nccl", rank=0, world_size=1)
    easydist_setup(backend="torch", device="cuda")
    set_device_mesh([0], ["spmd0"])
    a = run(False)
    b = run(True)
    for i, ((l0, c0), (l1, c1)) in enumerate(zip(a, b)):
        print(f"step {i}: eagerC loss={l0:.5f} csum={c0:.3f} | "
              f"graph loss={l1:.5f} csum={c1:.3f}")
    dist.destroy_process_group()
