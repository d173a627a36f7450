"""Shared utilities (meta conversion, signatures, attribute helpers, timing)."""
import hashlib
from contextlib import contextmanager
from functools import reduce

import torch
import torch.utils._pytree as pytree


def rgetattr(obj, path: str):
    return reduce(getattr, path.split("."), obj)


def rsetattr(obj, path: str, value):
    parts = path.split(".")
    for p in parts[:-1]:
        obj = getattr(obj, p)
    setattr(obj, parts[-1], value)


def to_meta(x):
    """Map tensors to meta tensors; leave everything else alone."""
    if isinstance(x, torch.Tensor):
        return x.detach().to(device="meta")
    return x


def tree_to_meta(tree):
    return pytree.tree_map(to_meta, tree)


def get_input_signature(*args, **kwargs):
    """Hash the meta-structure of the inputs (shape/dtype signature)."""

    def describe(x):
        if isinstance(x, torch.Tensor):
            return ("T", tuple(x.shape), str(x.dtype), tuple(x.stride()))
        return repr(x)

    flat, spec = pytree.tree_flatten([args, kwargs])
    sig = repr([describe(x) for x in flat]) + repr(spec)
    return hashlib.sha256(sig.encode("utf-8")).hexdigest()


@contextmanager
def _enable_compile():
    """Make torch's ``is_compiling`` return True so optimizers trace whole.

    Reference behavior: easydist/torch/utils.py:196-215. The optimizer code
    guards .item() / host sync behind ``torch.compiler.is_compiling()``; while
    tracing the train step whole-graph we need the compiled behavior.
    """

    def f_true():
        return True

    funcs = []
    for mod, name in [(torch._utils, "is_compiling"), (torch._dynamo, "is_compiling"),
                      (torch.compiler, "is_compiling")]:
        f = getattr(mod, name, None)
        if f is not None and callable(f):
            funcs.append(f)
    # dedup by code object
    seen, uniq = set(), []
    for f in funcs:
        if id(f.__code__) not in seen:
            seen.add(id(f.__code__))
            uniq.append(f)
    origs = [f.__code__ for f in uniq]
    for f in uniq:
        f.__code__ = f_true.__code__
    try:
        yield
    finally:
        for f, o in zip(uniq, origs):
            f.__code__ = o


@contextmanager
def _rematerialize_optimizer(opt, named_states, params):
    """Swap the optimizer's state dict for traced state tensors.

    Reference behavior: easydist/torch/utils.py:160-186.
    """
    if opt is None:
        yield
        return
    orig_states = dict(opt.state)
    orig_params = {i: g["params"] for i, g in enumerate(opt.param_groups)}
    # map the new param tensors in (by name order of `params`)
    name_list = list(params.keys())
    flat_params = [params[n] for n in name_list]
    # single param-group optimizers cover all our model cases; multi-group
    # optimizers keep their group structure by matching identity order
    pos = 0
    for g in opt.param_groups:
        n = len(g["params"])
        g["params"] = flat_params[pos:pos + n]
        pos += n
    opt.state.clear()
    for n, p in zip(name_list, flat_params):
        if n in named_states:
            opt.state[p] = named_states[n]
    try:
        yield
    finally:
        for i, g in enumerate(opt.param_groups):
            g["params"] = orig_params[i]
        opt.state.clear()
        opt.state.update(orig_states)
