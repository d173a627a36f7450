"""Actor helpers that stand up torch.distributed environments.

Capability parity with reference ``easydist/symphonia/torch_actor.py``
(a Ray actor exporting MASTER_ADDR/RANK env, lines 1-64). Ray is not in
this image, so the Ray path is lazy-imported and a multiprocessing
launcher provides the same capability locally (one process per GPU,
RCCL rendezvous over 127.0.0.1).
"""
from __future__ import annotations

import multiprocessing as mp
import os
from typing import Callable, List, Optional


class TorchDistActor:
    """Configure this process as rank `rank` of a torch.distributed job.

    Usable as a Ray actor (`ray.remote(TorchDistActor)`) or directly."""

    def __init__(self, rank: int, world_size: int,
                 master_addr: str = "127.0.0.1",
                 master_port: int = 29500):
        self.rank = rank
        self.world_size = world_size
        os.environ["MASTER_ADDR"] = master_addr
        os.environ["MASTER_PORT"] = str(master_port)
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)

    def init_process_group(self, backend: Optional[str] = None):
        import torch
        import torch.distributed as dist
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(self.rank)
        dist.init_process_group(backend=backend, rank=self.rank,
                                world_size=self.world_size)
        return True

    def run(self, fn: Callable, *args, **kwargs):
        return fn(*args, **kwargs)


def _worker(rank, world_size, port, fn, args):
    actor = TorchDistActor(rank, world_size, master_port=port)
    actor.init_process_group()
    fn(*args)


def launch_actors(fn: Callable, world_size: int, args: tuple = (),
                  port: int = 29510) -> List[mp.Process]:
    """Spawn world_size processes, each dist-initialized, running fn."""
    ctx = mp.get_context("spawn")
    procs = []
    for r in range(world_size):
        p = ctx.Process(target=_worker, args=(r, world_size, port, fn,
                                              args))
        p.start()
        procs.append(p)
    return procs
