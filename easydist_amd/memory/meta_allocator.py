"""Swap-in HIP pluggable allocator + profiling control.

Capability parity with reference ``easydist/torch/meta_allocator.py`` and
``easydist/torch/cuda/mem_allocator.py`` (init_meta_allocator /
swap_to_profiling_allocator, lines 25-59): loads the in-tree
``_mem_alloc`` extension (built by setup.py — the .so travels with the
repo snapshot) and installs it as the process allocator through
``torch.cuda.memory.CUDAPluggableAllocator`` (HIPified on ROCm).
"""
from __future__ import annotations

import logging
import os

import torch

logger = logging.getLogger(__name__)

PASSTHROUGH, PROFILE, RUNTIME = 0, 1, 2

_ctl = None          # pybind control module
_installed = False


def _ext_path():
    d = os.path.dirname(__file__)
    if not os.path.isdir(d):
        return None
    for name in os.listdir(d):
        if name.startswith("_mem_alloc") and name.endswith(".so"):
            return os.path.join(d, name)
    return None


def load_allocator_ext():
    global _ctl
    if _ctl is not None:
        return _ctl
    path = _ext_path()
    if path is None:
        return None
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "easydist_amd.memory._mem_alloc", path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _ctl = mod
    return _ctl


def init_meta_allocator() -> bool:
    """Install ed_malloc/ed_free as the process CUDA(HIP) allocator.

    Must run before the first device allocation
    (torch.cuda.memory.change_current_allocator constraint).
    """
    global _installed
    if _installed:
        return True
    if not torch.cuda.is_available():
        return False
    ctl = load_allocator_ext()
    if ctl is None:
        logger.warning("memory/_mem_alloc*.so not built: memory planning "
                       "disabled (run __graft_entry__.build())")
        return False
    path = _ext_path()
    alloc = torch.cuda.memory.CUDAPluggableAllocator(path, "ed_malloc",
                                                     "ed_free")
    torch.cuda.memory.change_current_allocator(alloc)
    ctl.set_mode(PASSTHROUGH)
    _installed = True
    logger.info("installed easydist_amd pluggable HIP allocator")
    return True


def allocator_installed() -> bool:
    return _installed


def ctl():
    assert _ctl is not None, "allocator extension not loaded"
    return _ctl
