"""Pipeline split markers: custom ops + annotation API + trace patcher.

Capability parity with reference ``easydist/torch/experimental/pp/
split_utils.py`` (fw_bw_split/step_split custom ops, lines 66-157;
SplitPatcher 217-303) and ``compile_pipeline.py`` (annotate_split_points
51-78, split_into_equal_size 81-230). Re-designed: one marker op
``easydist_amd::pp_split(x, idx, is_backward)`` whose autograd emits the
matching backward marker, so the whole-step trace carries both boundary
sets; the optimizer boundary is marked by routing every grad through
``easydist_amd::step_split`` from a patched ``Optimizer.step``.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional, Set

import torch

logger = logging.getLogger(__name__)

LOSS_BOUNDARY = -1   # idx of the fwd-end / bwd-start marker

lib = torch.library.Library("easydist_amd", "FRAGMENT")
lib.define("pp_split(Tensor x, int idx, bool is_backward) -> Tensor")
lib.define("step_split(Tensor g) -> Tensor")


def _pp_split_impl(x, idx, is_backward):
    return x.clone()


def _step_split_impl(g):
    return g.clone()


for backend in ("CPU", "CUDA"):
    lib.impl("pp_split", _pp_split_impl, backend)
    lib.impl("step_split", _step_split_impl, backend)


@torch.library.register_fake("easydist_amd::pp_split")
def _pp_split_fake(x, idx, is_backward):
    return x.new_empty(tuple(x.shape))


@torch.library.register_fake("easydist_amd::step_split")
def _step_split_fake(g):
    return g.new_empty(tuple(g.shape))


def _pp_split_backward(ctx, grad):
    g = torch.ops.easydist_amd.pp_split(grad, ctx.idx, True)
    return g, None, None


def _pp_split_setup(ctx, inputs, output):
    ctx.idx = inputs[1]


torch.library.register_autograd("easydist_amd::pp_split", _pp_split_backward,
                                setup_context=_pp_split_setup)


# ---------------------------------------------------------------- counters ---
class _SplitState:
    """Per-trace marker bookkeeping (reset by SplitPatcher.__enter__)."""
    boundary_counter = 0


def _next_boundary() -> int:
    i = _SplitState.boundary_counter
    _SplitState.boundary_counter += 1
    return i


def _mark_tree(out, idx: int):
    import torch.utils._pytree as pytree
    flat, spec = pytree.tree_flatten(out)
    marked = [torch.ops.easydist_amd.pp_split(v, idx, False)
              if isinstance(v, torch.Tensor) and v.is_floating_point()
              else v
              for v in flat]
    return pytree.tree_unflatten(marked, spec)


# ---------------------------------------------------------- annotation API ---
def annotate_split_points(module: torch.nn.Module, spots: Set[str]):
    """Insert a pipeline boundary AFTER each named submodule's forward.

    reference: compile_pipeline.py:51-78."""
    for qualname in sorted(spots):
        sub = module.get_submodule(qualname)
        if getattr(sub, "_ed_pp_annotated", False):
            continue
        sub._ed_pp_annotated = True

        def hook(mod, args, out):
            if not _SplitState.active:
                return out
            return _mark_tree(out, _next_boundary())
        sub.register_forward_hook(hook)


def split_into_equal_size(nstages: int):
    """Return a callable(module) -> module that annotates ~equal-parameter
    split points. reference: compile_pipeline.py:81-230."""
    def annotate(module: torch.nn.Module) -> torch.nn.Module:
        sizes = []
        for name, sub in module.named_children():
            n = sum(p.numel() for p in sub.parameters())
            sizes.append((name, n, sub))
        # flatten ModuleList children one level for finer granularity
        flat = []
        for name, n, sub in sizes:
            if isinstance(sub, (torch.nn.ModuleList, torch.nn.Sequential)):
                for cname, csub in sub.named_children():
                    flat.append((f"{name}.{cname}",
                                 sum(p.numel() for p in csub.parameters())))
            else:
                flat.append((name, n))
        total = sum(n for _, n in flat)
        target = total / nstages
        spots, acc, stage = set(), 0, 0
        for name, n in flat:
            acc += n
            if acc >= target * (stage + 1) and stage < nstages - 1:
                spots.add(name)
                stage += 1
        annotate_split_points(module, spots)
        return module
    return annotate


# -------------------------------------------------------------- SplitPatcher -
class SplitPatcher:
    """Context manager active during tracing: resets the boundary counter,
    enables the forward hooks, marks the loss boundary by patching
    ``Tensor.backward`` and routes grads through ``step_split`` by patching
    ``Optimizer.step``. reference: split_utils.py:217-303."""

    def __init__(self, module: Optional[torch.nn.Module], opt):
        self.module = module
        self.opt = opt

    def __enter__(self):
        _SplitState.boundary_counter = 0
        _SplitState.active = True
        self._orig_backward = torch.Tensor.backward
        # patch the CONCRETE optimizer class: subclasses (Adam, SGD, ...)
        # override step, so patching the torch.optim.Optimizer base would
        # never fire
        self._opt_cls = type(self.opt) if self.opt is not None else None
        self._orig_step = self._opt_cls.step if self._opt_cls else None

        orig_backward = self._orig_backward

        def patched_backward(t, *a, **kw):
            marked = torch.ops.easydist_amd.pp_split(t, LOSS_BOUNDARY, False)
            return orig_backward(marked, *a, **kw)

        orig_step = self._orig_step

        def patched_step(opt_self, *a, **kw):
            for group in opt_self.param_groups:
                for p in group["params"]:
                    if p.grad is not None:
                        p.grad = torch.ops.easydist_amd.step_split(p.grad)
            return orig_step(opt_self, *a, **kw)

        torch.Tensor.backward = patched_backward
        if self._opt_cls is not None:
            self._opt_cls.step = patched_step
        return self

    def __exit__(self, *exc):
        torch.Tensor.backward = self._orig_backward
        if self._opt_cls is not None:
            self._opt_cls.step = self._orig_step
        _SplitState.active = False
        return False


_SplitState.active = False
