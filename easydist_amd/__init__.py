"""easydist_amd: an MI355X-native auto-parallel training compiler.

A from-scratch framework with the capabilities of alibaba/easydist
(see SURVEY.md): trace an unmodified PyTorch train step whole-graph
(fwd+bwd+optimizer), discover per-op SPMD rules by sharded execution
(ShardCombine), solve a MILP for the cheapest strategy under an
xGMI/CDNA4 cost model, rewrite the graph with RCCL collectives, and run
it under hipGraph with hand-written HIP kernels on the hot ops.
"""
import logging

from . import config as mdconfig
from .compiler.api import easydist_compile, register_parallel_method
from .parallel.device_mesh import (NDDeviceMesh, get_device_mesh,
                                   set_device_mesh)

__version__ = "0.2.0"   # salts the strategy cache

logger = logging.getLogger("easydist_amd")


def easydist_setup(backend: str = "torch", device: str = "cuda",
                   allow_tf32: bool = True):
    """One-call environment setup (reference: easydist/__init__.py:21).

    On ROCm: enables hipblaslt TF32-equivalent paths where applicable, caps
    the device connection count so collective launch order is deterministic
    (needed by the static memory planner), and installs logging.
    """
    import os

    import torch

    logging.basicConfig(
        level=getattr(logging, mdconfig.log_level.upper(), logging.INFO),
        format="[%(asctime)s %(name)s %(levelname)s] %(message)s")
    mdconfig.easydist_device = device
    if backend != "torch":
        raise NotImplementedError(
            "easydist_amd is torch-only by design (SURVEY.md §7): the "
            "reference's jax/tvm platform indirection is collapsed")
    if allow_tf32:
        torch.backends.cuda.matmul.allow_tf32 = True
        torch.backends.cudnn.allow_tf32 = True
    # deterministic collective/compute interleaving (reference:
    # easydist/torch/__init__.py:42-56); the HIP analog governs stream
    # connection multiplexing
    os.environ.setdefault("CUDA_DEVICE_MAX_CONNECTIONS", "1")
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    if mdconfig.enable_memory_opt and device == "cuda":
        # must happen before the first device allocation
        from .memory import init_meta_allocator
        init_meta_allocator()
    return True


__all__ = [
    "easydist_setup", "easydist_compile", "register_parallel_method",
    "set_device_mesh", "get_device_mesh", "NDDeviceMesh", "mdconfig",
]
