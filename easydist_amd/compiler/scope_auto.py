"""scope_auto: mark code regions, extract them as call_module subgraphs.

Capability parity with reference ``easydist/scope_auto/``
(scope_marker.py:30-163, build_scope_modules.py:49-308): a decorated
function's trace is bracketed by identity marker ops; a post-pass lifts
each bracketed region into its own submodule (the reference's multi-mesh
preparation step).
"""
from __future__ import annotations

import functools
import logging
from typing import Dict, List

import torch
import torch.fx as fx
import torch.utils._pytree as pytree

logger = logging.getLogger(__name__)

lib = torch.library.Library("easydist_amd", "FRAGMENT")
lib.define("scope_enter(Tensor x, int scope_id) -> Tensor")
lib.define("scope_exit(Tensor x, int scope_id) -> Tensor")

for _backend in ("CPU", "CUDA"):
    lib.impl("scope_enter", lambda x, scope_id: x.clone(), _backend)
    lib.impl("scope_exit", lambda x, scope_id: x.clone(), _backend)


@torch.library.register_fake("easydist_amd::scope_enter")
def _se_fake(x, scope_id):
    return x.new_empty(tuple(x.shape))


@torch.library.register_fake("easydist_amd::scope_exit")
def _sx_fake(x, scope_id):
    return x.new_empty(tuple(x.shape))


_SCOPE_COUNTER = [0]
_SCOPE_NAMES: Dict[int, str] = {}


def scope_marker(name: str = ""):
    """Decorator: wrap the callable's tensor inputs/outputs in scope
    markers so the traced graph carries the region boundary."""

    def deco(fn):
        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            sid = _SCOPE_COUNTER[0]
            _SCOPE_COUNTER[0] += 1
            _SCOPE_NAMES[sid] = name or fn.__name__
            flat, spec = pytree.tree_flatten((args, kwargs))
            flat = [torch.ops.easydist_amd.scope_enter(v, sid)
                    if isinstance(v, torch.Tensor) else v for v in flat]
            a2, k2 = pytree.tree_unflatten(flat, spec)
            out = fn(*a2, **k2)
            oflat, ospec = pytree.tree_flatten(out)
            oflat = [torch.ops.easydist_amd.scope_exit(v, sid)
                     if isinstance(v, torch.Tensor) else v for v in oflat]
            return pytree.tree_unflatten(oflat, ospec)
        return wrapper
    return deco


ENTER = torch.ops.easydist_amd.scope_enter.default
EXIT = torch.ops.easydist_amd.scope_exit.default


def build_scope_modules(gm: fx.GraphModule) -> fx.GraphModule:
    """Extract every scope region into a named call_module submodule."""
    nodes = list(gm.graph.nodes)
    # group marker nodes by scope id
    enters: Dict[int, List[fx.Node]] = {}
    exits: Dict[int, List[fx.Node]] = {}
    for n in nodes:
        if n.op == "call_function" and n.target is ENTER:
            enters.setdefault(n.args[1], []).append(n)
        elif n.op == "call_function" and n.target is EXIT:
            exits.setdefault(n.args[1], []).append(n)

    pos = {n: i for i, n in enumerate(nodes)}
    for sid in sorted(enters):
        if sid not in exits:
            continue
        first = min(pos[n] for n in enters[sid])
        last = max(pos[n] for n in exits[sid])
        seg = [n for n in nodes[first:last + 1]
               if n.op == "call_function" and n.target not in (ENTER, EXIT)]
        if not seg:
            continue
        seg_set = set(seg)
        alias = {m: m.args[0] for m in enters[sid] + exits[sid]}

        sub_graph = fx.Graph()
        env: Dict[fx.Node, fx.Node] = {}
        inputs: List[fx.Node] = []

        def resolve(n):
            while n in alias:
                n = alias[n]
            return n

        def lookup(n):
            n = resolve(n)
            if n in env:
                return env[n]
            ph = sub_graph.placeholder(n.name)
            if "val" in n.meta:
                ph.meta["val"] = n.meta["val"]
            env[n] = ph
            inputs.append(n)
            return ph

        for n in seg:
            env[n] = sub_graph.node_copy(n, lookup)
        # region outputs: exit-marker inputs + any value used outside
        out_vals: List[fx.Node] = []
        for x in exits[sid]:
            v = resolve(x.args[0])
            if v in seg_set and v not in out_vals:
                out_vals.append(v)
        for n in seg:
            for u in n.users:
                if u not in seg_set and resolve(u) not in seg_set \
                        and n not in out_vals:
                    out_vals.append(n)
        sub_graph.output(tuple(env[resolve(v)] for v in out_vals))

        mod_name = f"scope_{_SCOPE_NAMES.get(sid, sid)}_{sid}"
        gm.add_submodule(mod_name, fx.GraphModule(gm, sub_graph))
        with gm.graph.inserting_before(seg[0]):
            call = gm.graph.call_module(mod_name, tuple(inputs))
        # strip markers FIRST so outside users point at raw region values,
        # then reroute those users onto the call_module outputs
        for x in exits[sid] + enters[sid]:
            x.replace_all_uses_with(resolve(x))
        for k, v in enumerate(out_vals):
            with gm.graph.inserting_after(call):
                item = gm.graph.call_function(
                    __import__("operator").getitem, (call, k))
            for u in list(v.users):
                if u not in seg_set and u is not item and u is not call:
                    u.replace_input_with(v, item)
        for n in reversed(seg):
            if not n.users:
                gm.graph.erase_node(n)
        for x in exits[sid] + enters[sid]:
            if not x.users:
                gm.graph.erase_node(x)
    gm.graph.eliminate_dead_code()
    gm.graph.lint()
    gm.recompile()
    return gm
