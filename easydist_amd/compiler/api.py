"""easydist_compile: the one-decorator user API.

Capability parity with reference ``easydist/torch/api.py``
(easydist_compile / CompiledFuncWrapper / register_parallel_method,
lines 39-256): lazy compile keyed by input signature, parallel_mode
auto|ddp|zero2|zero3|pp + plugin registry, hipGraph capture on by default.
"""
from __future__ import annotations

import logging
from functools import update_wrapper
from typing import Callable, Dict, Optional

import torch
import torch.utils._pytree as pytree

from .. import config as mdconfig
from ..utils import get_input_signature

logger = logging.getLogger(__name__)

PARALLEL_EXTENSION: Dict[str, Callable] = {}


def register_parallel_method(parallel_mode: str, compiler_func=None):
    def wrapper(fn):
        PARALLEL_EXTENSION[parallel_mode] = fn
        logger.info("registered parallel method [%s]", parallel_mode)
        return fn
    return wrapper(compiler_func) if compiler_func else wrapper


def _find_module_opt(args, kwargs):
    module, opt = None, None
    flat = list(args) + list(kwargs.values())
    for a in flat:
        if isinstance(a, torch.nn.Module) and module is None:
            module = a
        if isinstance(a, torch.optim.Optimizer) and opt is None:
            opt = a
    return module, opt


class CompiledFuncWrapper:
    def __init__(self, func, parallel_mode="auto", tracing_mode="fake",
                 cuda_graph=True, compile_only=False, memory_opt=False,
                 **compile_kwargs):
        update_wrapper(self, func)
        self.original_func = func
        self.parallel_mode = parallel_mode
        self.tracing_mode = tracing_mode
        self.enable_cuda_graph = cuda_graph and mdconfig.enable_hip_graph
        self.memory_opt = memory_opt or mdconfig.enable_memory_opt
        if self.memory_opt:
            self.enable_cuda_graph = False   # plan playback is eager-mode
        self.compile_only = compile_only
        self.compile_kwargs = compile_kwargs
        self.compiled: Dict[str, object] = {}   # input signature -> runtime

    def _compile(self, args, kwargs):
        module, opt = _find_module_opt(args, kwargs)
        if self.parallel_mode == "auto":
            from .compile_auto import _compile_auto
            return _compile_auto(self.original_func, self.tracing_mode, args,
                                 kwargs, module, opt)
        if self.parallel_mode in ("ddp", "zero2", "zero3"):
            from .compile_dp import _compile_dp
            return _compile_dp(self.original_func, self.parallel_mode,
                               self.tracing_mode, args, kwargs, module, opt)
        if self.parallel_mode == "pp":
            from ..parallel.pp.api import _compile_pp
            return _compile_pp(self.original_func, self.tracing_mode, args,
                               kwargs, module, opt, **self.compile_kwargs)
        if self.parallel_mode in PARALLEL_EXTENSION:
            return PARALLEL_EXTENSION[self.parallel_mode](
                self.original_func, self.tracing_mode, args, kwargs, module,
                opt, **self.compile_kwargs)
        raise NotImplementedError(self.parallel_mode)

    def __call__(self, *args, **kwargs):
        sig = get_input_signature(*args, **kwargs)
        if sig not in self.compiled:
            logger.info("compiling for new input signature %s...", sig[:12])
            self.compiled[sig] = self._compile(args, kwargs)
            if self.compile_only:
                return self.compiled[sig]
        runtime = self.compiled[sig]
        if hasattr(runtime, "run_pipeline"):      # pp runtime
            return runtime.run_pipeline(args, kwargs)
        module, opt = _find_module_opt(args, kwargs)
        params = dict(module.named_parameters()) if module else {}
        buffers = dict(module.named_buffers()) if module else {}
        flat_inputs, _ = pytree.tree_flatten(
            (params, buffers, runtime.compile_named_states
             if hasattr(runtime, "compile_named_states") else {},
             args, kwargs))
        # note: state positions come from the runtime's persistent buffers
        # after the first call; the flat list only seeds them once.
        flat_inputs = self._reflatten(runtime, params, buffers, args, kwargs)
        if self.memory_opt and torch.cuda.is_available() \
                and hasattr(runtime, "run_planned"):
            outs = runtime.run_planned(flat_inputs)
        elif (self.enable_cuda_graph and torch.cuda.is_available()):
            outs = runtime.run_graph(flat_inputs)
        else:
            outs = runtime.run(flat_inputs)
        return self._unflatten_ret(runtime, outs)

    def _reflatten(self, runtime, params, buffers, args, kwargs):
        ns = getattr(runtime, "init_named_states", None) or {}
        flat, _ = pytree.tree_flatten((params, buffers, ns, args, kwargs))
        return flat

    def _unflatten_ret(self, runtime, flat_outs):
        spec = runtime.meta.get("out_spec") if hasattr(runtime, "meta") else None
        if spec is None:
            return flat_outs
        full = pytree.tree_unflatten(list(flat_outs), spec)
        # (params, buffers, states, grads, ret) -> user sees ret
        return full[4]

    # ------------------------------------------------- state access APIs ----
    def get_state(self):
        rts = list(self.compiled.values())
        assert rts, "not compiled yet"
        return rts[0].get_state()

    def parameters(self):
        rts = list(self.compiled.values())
        return rts[0].named_parameters().values()

    def named_parameters(self):
        rts = list(self.compiled.values())
        return rts[0].named_parameters()


def easydist_compile(func=None, parallel_mode="auto", tracing_mode="fake",
                     cuda_graph=True, use_hint=False,
                     max_solver_time=float("inf"), compile_only=False,
                     memory_opt=False, **compile_kwargs):
    # reference semantics (sharding_interpreter.py:270): use_hint shrinks
    # giant-op probes during discovery; our discovery ALWAYS probes at
    # discovery_max_dim-shrunk shapes (sharding_interpreter._shrink_map),
    # so the flag is accepted for API parity and is a no-op
    mdconfig.use_hint = use_hint
    mdconfig.max_seconds_same_incumbent = max_solver_time

    def deco(fn):
        return CompiledFuncWrapper(fn, parallel_mode, tracing_mode,
                                   cuda_graph, compile_only, memory_opt,
                                   **compile_kwargs)
    return deco(func) if func else deco
