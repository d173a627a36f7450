"""Deferred materialization of meta/fake parameters.

Capability parity with reference ``easydist/torch/init_helper.py``
(init_contiguous_buf / CpuModuleInitHelper / SetParaInitHelper,
lines 23-166): big models are built on the meta device, compiled, and
only the LOCAL shard of each parameter is materialized on the MI355X —
288 GB HBM3E per GPU means the shard fits where the global tensor often
would not on the reference's hardware.
"""
from __future__ import annotations

import logging
from typing import Callable, Dict, Optional

import torch

logger = logging.getLogger(__name__)


class InitHelper:
    """Strategy object producing real tensors for fake/meta params."""

    def materialize(self, name: str, t: torch.Tensor,
                    device: str) -> torch.Tensor:
        raise NotImplementedError


class ZeroInitHelper(InitHelper):
    def materialize(self, name, t, device):
        return torch.zeros(tuple(t.shape), dtype=t.dtype, device=device)


class RandomInitHelper(InitHelper):
    def __init__(self, std: float = 0.02, seed: Optional[int] = None):
        self.std = std
        self.seed = seed

    def materialize(self, name, t, device):
        g = None
        if self.seed is not None:
            g = torch.Generator(device=device)
            # per-tensor deterministic seed so every rank agrees
            g.manual_seed(self.seed + (hash(name) & 0xFFFF))
        if t.is_floating_point():
            return torch.empty(tuple(t.shape), dtype=t.dtype,
                               device=device).normal_(0, self.std,
                                                      generator=g)
        return torch.zeros(tuple(t.shape), dtype=t.dtype, device=device)


class CpuModuleInitHelper(InitHelper):
    """Copy values from a CPU twin of the module (reference:
    init_helper.py:88-117)."""

    def __init__(self, cpu_module: torch.nn.Module):
        self.state = {**dict(cpu_module.named_parameters()),
                      **dict(cpu_module.named_buffers())}

    def materialize(self, name, t, device):
        src = self.state.get(name)
        assert src is not None, f"no CPU value for {name}"
        return src.detach().to(device)


class ResetParametersInitHelper(InitHelper):
    """Call each submodule's reset_parameters() on a real-device twin."""

    def __init__(self, module_factory: Callable[[], torch.nn.Module]):
        self.factory = module_factory
        self._state: Optional[Dict[str, torch.Tensor]] = None

    def materialize(self, name, t, device):
        if self._state is None:
            m = self.factory()
            self._state = {**dict(m.named_parameters()),
                           **dict(m.named_buffers())}
        return self._state[name].detach().to(device)


def materialize_module(module: torch.nn.Module, helper: InitHelper,
                       device: str = "cuda") -> torch.nn.Module:
    """Replace every meta/fake parameter and buffer with a real tensor."""
    for name, p in list(module.named_parameters()):
        if p.device.type == "meta":
            new = helper.materialize(name, p, device)
            _set_by_qualname(module, name,
                             torch.nn.Parameter(new, p.requires_grad))
    for name, b in list(module.named_buffers()):
        if b.device.type == "meta":
            _set_by_qualname(module, name, helper.materialize(name, b,
                                                              device))
    return module


def _set_by_qualname(module, qualname: str, value):
    parts = qualname.split(".")
    owner = module
    for p in parts[:-1]:
        owner = getattr(owner, p)
    if isinstance(value, torch.nn.Parameter):
        owner._parameters[parts[-1]] = value
    else:
        if parts[-1] in owner._buffers:
            owner._buffers[parts[-1]] = value
        else:
            setattr(owner, parts[-1], value)
