"""Whole-graph tracing: fwd + bwd + optimizer step as ONE fx graph.

Capability parity with reference ``easydist/torch/compile.py``
(stateless_func / ed_compile_func, lines 25-120), re-designed for
torch-2.10/ROCm: fused-optimizer decomposition happens at trace time and the
resulting graph is made pure by the functionalize pass.
"""
from __future__ import annotations

import logging
from contextlib import nullcontext
from functools import partial
from typing import Callable, Optional

import torch
import torch.utils._pytree as pytree
from torch._subclasses.fake_tensor import FakeTensor
from torch.fx.experimental.proxy_tensor import make_fx
from torch.nn.utils import stateless

from ..utils import _enable_compile, _rematerialize_optimizer
from .decomp import EASYDIST_DECOMP_TABLE

logger = logging.getLogger(__name__)


def stateless_func(func, module, opt, params, buffers, named_states, args,
                   kwargs):
    """Make the user's train_step pure in (params, buffers, states)."""
    ctx1 = (stateless._reparametrize_module(module, {**params, **buffers},
                                            tie_weights=True)
            if module is not None else nullcontext())
    ctx2 = (_rematerialize_optimizer(opt, named_states, params)
            if opt is not None else nullcontext())
    with ctx1, ctx2:
        ret = func(*args, **kwargs)
    grads = {k: v.grad for k, v in params.items()}
    return params, buffers, named_states, grads, ret


def warmup_optimizer(module: torch.nn.Module, opt) -> dict:
    """Create optimizer states before tracing by running one zero-grad step.

    (reference behavior: compile.py:50-66 — the warm-up step materializes
    exp_avg etc.; the step counter is rewound by one so the traced graph's
    increment reproduces the true first step when executed on live state.)
    """
    named_states = {}
    if opt is None:
        return named_states
    params = dict(module.named_parameters())
    fresh = {n for n, p in params.items() if not opt.state.get(p)}
    if fresh:
        with torch.no_grad():
            psnap = {n: p.detach().clone() for n, p in params.items()}
            # pre-existing states (resumed optimizer) must survive the
            # fake step untouched
            ssnap = {n: {k: (v.detach().clone() if torch.is_tensor(v)
                             else v)
                         for k, v in opt.state[p].items()}
                     for n, p in params.items() if n not in fresh}
            for p in params.values():
                if p.grad is None:
                    p.grad = torch.zeros_like(p)
        opt.step()
        opt.zero_grad(True)
        with torch.no_grad():
            # the warm-up step is fake: undo any param drift (weight
            # decay etc.)
            for n, p in params.items():
                p.copy_(psnap[n])
            for n, p in params.items():
                st = opt.state.get(p)
                if not st:
                    continue
                if n in ssnap:
                    for k, v in ssnap[n].items():
                        if torch.is_tensor(v) and torch.is_tensor(st.get(k)):
                            st[k].copy_(v)
                        else:
                            st[k] = v
                else:
                    # freshly materialized: the fake step left wd/momentum
                    # residue in the moment buffers (g was 0 but wd*p was
                    # not) — reset to true step-0 state
                    for k, v in list(st.items()):
                        if k == "step":
                            st[k] = v - 1
                        elif torch.is_tensor(v):
                            v.zero_()
    for n, p in params.items():
        if p in opt.state:
            named_states[n] = dict(opt.state[p])
    return named_states


def ed_compile_func(func: Callable, tracing_mode: str, args, kwargs,
                    module: Optional[torch.nn.Module], opt,
                    split_patcher_ctx=None, decomp_table=None):
    """Trace `func(module, opt, *args)` into one whole-step fx graph.

    Returns (params, buffers, named_states, traced_graph).
    `decomp_table` defaults to EASYDIST_DECOMP_TABLE; the manual-DP path
    passes {} so fused/foreach optimizer ops stay whole in the graph.
    """
    params, buffers = {}, {}
    if module is not None:
        params = dict(module.named_parameters())
        buffers = dict(module.named_buffers())
    named_states = warmup_optimizer(module, opt) if opt is not None else {}

    if decomp_table is None:
        decomp_table = EASYDIST_DECOMP_TABLE
    ctx = split_patcher_ctx if split_patcher_ctx is not None else nullcontext()
    with _enable_compile(), ctx:
        traced_graph = make_fx(partial(stateless_func, func, module, opt),
                               tracing_mode=tracing_mode,
                               decomposition_table=decomp_table,
                               _allow_non_fake_inputs=False)(
                                   params, buffers, named_states, args, kwargs)
    traced_graph.graph.eliminate_dead_code()
    traced_graph.recompile()
    return params, buffers, named_states, traced_graph
