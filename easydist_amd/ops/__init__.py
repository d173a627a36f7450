"""Hand-written HIP/CDNA4 kernels, exposed as torch custom ops.

The compute path of the framework (SURVEY.md §7 phase 5): partitioned hot
ops in the lowered graph run as gfx950 kernels — MFMA-tiled GEMM, flash
attention, LayerNorm fwd/bwd, fused multi-tensor Adam, fused cross-entropy.
Plain large GEMMs go to hipBLASLt via aten (torch.mm); everything listed
here is a hand kernel.

Loading contract: on a GPU box the in-tree extension MUST load — a missing
.so raises instead of silently falling back to aten (the CPU fallbacks are
for CPU-only hosts and numerics tests only). Set EASYDIST_USE_HIP_KERNELS=0
to explicitly opt out (development only).
"""
from __future__ import annotations

import logging
import os

import torch

from .. import config as mdconfig

logger = logging.getLogger(__name__)

_EXT = None
_EXT_ERR = None


def _ext_path():
    d = os.path.dirname(__file__)
    for name in os.listdir(d) if os.path.isdir(d) else []:
        if name.startswith("_hip_ops") and name.endswith(".so"):
            return os.path.join(d, name)
    return None


def load_extension():
    """Load the in-tree HIP extension (built by setup.py / __graft_entry__)."""
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    path = _ext_path()
    if path is None:
        _EXT_ERR = "easydist_amd/ops/_hip_ops*.so not built (run __graft_entry__.build())"
        return None
    try:
        torch.ops.load_library(path)
        _EXT = torch.ops.easydist_amd_hip
        logger.info("loaded HIP kernel extension: %s", path)
        return _EXT
    except Exception as e:  # pragma: no cover
        _EXT_ERR = f"failed to load {path}: {e}"
        return None


def hip_ops_available() -> bool:
    return load_extension() is not None


def require_hip_ops():
    """On a GPU box the HIP kernels are mandatory: fail loudly."""
    if not torch.cuda.is_available():
        return False
    if not mdconfig.use_hip_kernels:
        return False
    if load_extension() is None:
        raise RuntimeError(
            f"easydist_amd HIP kernel extension missing on a GPU host: "
            f"{_EXT_ERR}. Build it with `python __graft_entry__.py build` "
            f"or `python setup.py build_ext --inplace`.")
    return True


from . import attention, norms, optim, ce, gemm, moe_ops  # noqa: E402

__all__ = ["attention", "norms", "optim", "ce", "gemm", "moe_ops",
           "load_extension", "hip_ops_available", "require_hip_ops"]
