"""Multi-process test harness.

Capability parity with reference ``easydist/utils/testing/spawn.py``
(lines 71-280): fork N ranks, init torch.distributed, run the test body,
pickle failures back. CPU tests use gloo (world_size > 1 works without a
GPU); on an MI355X box the same harness runs over RCCL.
"""
from __future__ import annotations

import functools
import os
import traceback
from typing import Callable

import torch
import torch.multiprocessing as mp


def _worker(rank, fn, args, world_size, port, backend, q):
    try:
        import torch.distributed as dist
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        if backend == "nccl":
            # multi-rank-per-GPU (RCCL supports it): wrap on the visible
            # device count so the ws2 collective paths run on a 1-GPU box
            torch.cuda.set_device(rank % max(torch.cuda.device_count(), 1))
        dist.init_process_group(backend=backend, rank=rank,
                                world_size=world_size,
                                init_method=f"tcp://127.0.0.1:{port}")
        fn(*args)
        dist.destroy_process_group()
        q.put((rank, None))
    except Exception:
        q.put((rank, traceback.format_exc()))


def spawn(fn: Callable, args=(), world_size: int = 2, port: int = 29531,
          backend: str = None, timeout: float = 300.0):
    """Run fn on `world_size` ranks; raise on any failure."""
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = []
    for r in range(world_size):
        p = ctx.Process(target=_worker,
                        args=(r, fn, args, world_size, port, backend, q))
        p.start()
        procs.append(p)
    errors = []
    for _ in range(world_size):
        rank, err = q.get()
        if err is not None:
            errors.append((rank, err))
    for p in procs:
        p.join(timeout)
        if p.is_alive():
            p.terminate()
    if errors:
        msgs = "\n".join(f"--- rank {r} ---\n{e}" for r, e in errors)
        raise RuntimeError(f"spawned test failed:\n{msgs}")


class TorchMockDeviceMesh:
    """Single-process mock mesh for annotation tests (no dist init)."""

    def __init__(self, *shape):
        self.shape = list(shape)

    def size(self, dim=None):
        if dim is None:
            n = 1
            for s in self.shape:
                n *= s
            return n
        return self.shape[dim]


def init_single_process(backend: str = None, port: int = 29599):
    """init_process_group with world_size=1 (for local tests/bench)."""
    import torch.distributed as dist
    if dist.is_initialized():
        return
    backend = backend or ("nccl" if torch.cuda.is_available() else "gloo")
    port = int(os.environ.get("MASTER_PORT", port))   # env wins
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(port))
    dist.init_process_group(backend=backend, rank=0, world_size=1,
                            init_method=f"tcp://127.0.0.1:{port}")
