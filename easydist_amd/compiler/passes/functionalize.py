"""Graph canonicalization: make the traced whole-step graph pure.

Replaces the reference's copy_-preserving design plus its fix-pass family
(easydist/torch/passes/{fix_*.py, eliminate_detach.py}): we instead remove
every top-level ``aten.copy_(placeholder, src)``, rewire the graph outputs to
``src``, and record the placeholder<->output pairing as the state_io_map.
The runtime writes results back into persistent buffers, so the executed
graph stays side-effect free — which is what hipGraph capture and the static
memory planner want.
"""
from __future__ import annotations

import logging
import operator
from typing import Dict

import torch
import torch.fx as fx

logger = logging.getLogger(__name__)

aten = torch.ops.aten


def eliminate_detach(gm: fx.GraphModule) -> fx.GraphModule:
    """detach is a no-op in an already-traced joint graph."""
    for node in list(gm.graph.nodes):
        if node.op == "call_function" and node.target in (
                aten.detach.default, aten.alias.default, aten.lift_fresh_copy.default):
            node.replace_all_uses_with(node.args[0])
            gm.graph.erase_node(node)
    return gm


def defunctionalize_copies(gm: fx.GraphModule) -> Dict[str, str]:
    """Remove aten.copy_(dst, src) where dst is a placeholder (train-state
    writeback). Returns state_io_map: placeholder node name -> src node name.
    """
    io_map: Dict[str, str] = {}
    for node in list(gm.graph.nodes):
        if node.op != "call_function" or node.target is not aten.copy_.default:
            continue
        dst, src = node.args[0], node.args[1]
        while (isinstance(dst, fx.Node) and dst.op == "call_function"
               and dst.target is aten.copy_.default):
            dst = dst.args[0]
        if isinstance(dst, fx.Node) and dst.op == "placeholder":
            io_map[dst.name] = src.name if isinstance(src, fx.Node) else None
            node.replace_all_uses_with(src)
            gm.graph.erase_node(node)
        else:
            logger.warning("copy_ into non-placeholder %s kept in graph", dst)
    return io_map


def fix_inplace(gm: fx.GraphModule) -> fx.GraphModule:
    """Convert trivial remaining in-place aten ops to functional form."""
    for node in list(gm.graph.nodes):
        if node.op != "call_function":
            continue
        name = getattr(node.target, "_opname", "") or str(node.target)
        overload = getattr(node.target, "overloadpacket", None)
        base = getattr(overload, "__name__", "") if overload else ""
        if base.endswith("_") and base not in ("copy_", "set_"):
            func_name = base[:-1]
            func_pkt = getattr(aten, func_name, None)
            if func_pkt is None:
                continue
            # keep the same overload: add_.Tensor -> add.Tensor etc.
            ol_name = getattr(node.target, "_overloadname", "default")
            new_target = getattr(func_pkt, ol_name, None)
            if new_target is None:
                try:
                    new_target = func_pkt.default
                except Exception:
                    continue
            with gm.graph.inserting_after(node):
                new_node = gm.graph.call_function(new_target, node.args,
                                                  node.kwargs)
                new_node.meta = dict(node.meta)
            node.replace_all_uses_with(new_node)
            # later reads of the mutated arg must see the new value
            mutated = node.args[0]
            if isinstance(mutated, fx.Node):
                after = False
                for n in list(gm.graph.nodes):
                    if n is new_node:
                        after = True
                        continue
                    if after and mutated in n.all_input_nodes:
                        n.replace_input_with(mutated, new_node)
            gm.graph.erase_node(node)
    return gm


def fix_overload_binding(gm: fx.GraphModule) -> fx.GraphModule:
    """Repair traced nodes whose args don't bind to their overload.

    make_fx records dropout's mask draw as ``bernoulli.default(x, p)``
    (two positionals), but that overload's schema only takes ``self`` —
    executing the graph raises. Rebind to the ``.p`` overload."""
    import torch
    for n in gm.graph.nodes:
        if n.op == "call_function" \
                and n.target is torch.ops.aten.bernoulli.default \
                and len(n.args) == 2:
            n.target = torch.ops.aten.bernoulli.p
    return gm


def canonicalize(gm: fx.GraphModule):
    """Run the full pre-sharding pass stack. Returns (gm, state_io_map)."""
    eliminate_detach(gm)
    fix_inplace(gm)
    # AFTER fix_inplace: bernoulli_.float functionalizes to
    # bernoulli.default (the packet has no .float overload), which then
    # needs its p argument rebound to the .p overload
    fix_overload_binding(gm)
    io_map = defunctionalize_copies(gm)
    gm.graph.eliminate_dead_code()
    gm.recompile()
    return gm, io_map
