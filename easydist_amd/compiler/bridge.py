"""fx graph -> MetaGraph bridge.

Capability parity with reference ``easydist/torch/bridge.py``
(torch2meta_graph, lines 52-247): converts the canonicalized traced graph
plus discovered sharding info into the solver's MetaGraph, fills matmul-
family FLOPs for the CDNA4 roofline term, and derives the solver's output
constraints (state outputs follow their input placement; user-visible
returns replicate).
"""
from __future__ import annotations

import logging
import operator
from typing import Dict, List, Optional, Tuple

import torch
import torch.fx as fx
import torch.utils._pytree as pytree

from ..metashard.metair import MetaGraph, MetaNode, MetaVar

logger = logging.getLogger(__name__)

aten = torch.ops.aten


def _dtype_bytes(dt: torch.dtype) -> int:
    try:
        return dt.itemsize
    except Exception:
        return torch.empty((), dtype=dt).element_size()


def _flops_of(node: fx.Node) -> float:
    """FLOPs for matmul-family ops (drives the MFMA roofline cost)."""
    t = node.target
    def shape(i):
        a = node.args[i]
        v = a.meta.get("val") if isinstance(a, fx.Node) else None
        return tuple(v.shape) if isinstance(v, torch.Tensor) else None
    try:
        ed = torch.ops.easydist_amd
        if t in (ed.flash_attention.default,):
            q = shape(0)
            # QK^T + PV, causal halves the work
            return 2.0 * q[0] * q[1] * q[2] * q[2] * q[3] * 2 * 0.5
        if t in (ed.flash_attention_bwd.default,):
            q = shape(1)
            return 2.0 * q[0] * q[1] * q[2] * q[2] * q[3] * 5 * 0.5
        if t in (aten._scaled_dot_product_flash_attention.default
                 if hasattr(aten, "_scaled_dot_product_flash_attention")
                 else aten.mm.default,
                 aten._scaled_dot_product_flash_attention_for_cpu.default,):
            q = shape(0)
            if q and len(q) == 4:
                return 2.0 * q[0] * q[1] * q[2] * q[2] * q[3] * 2 * 0.5
        if t in (aten.mm.default,):
            a, b = shape(0), shape(1)
            return 2.0 * a[0] * a[1] * b[1]
        if t in (aten.addmm.default,):
            a, b = shape(1), shape(2)
            return 2.0 * a[0] * a[1] * b[1]
        if t in (aten.bmm.default,):
            a, b = shape(0), shape(1)
            return 2.0 * a[0] * a[1] * a[2] * b[2]
        if t in (aten.convolution.default,):
            out = node.meta.get("val")
            w = shape(1)
            if out is not None and w is not None:
                pos = 1
                for s in out.shape:
                    pos *= s
                k = 1
                for s in w[1:]:
                    k *= s
                return 2.0 * pos * k
        if t in (aten._scaled_dot_product_flash_attention.default,
                 aten._scaled_dot_product_efficient_attention.default,
                 aten.scaled_dot_product_attention.default):
            q = shape(0)
            if q is not None:
                b, h, s, d = q if len(q) == 4 else (1, *q)
                return 4.0 * b * h * s * s * d
    except Exception:
        pass
    return 0.0


def fx2meta_graph(gm: fx.GraphModule, sharding_info: Dict[str, Tuple],
                  state_io_map: Dict[str, str],
                  ret_names: Optional[set] = None):
    """Build the MetaGraph.

    Returns (meta_graph, output_constraints, var_of_fxnode) where
    var_of_fxnode maps fx node name -> MetaVar name (resolving getitem).
    """
    g = MetaGraph(gm.__class__.__name__)
    var_of: Dict[str, str] = {}          # fx node name -> var name
    meta_vars: Dict[str, MetaVar] = {}

    def var_for(node: fx.Node, val: torch.Tensor, suffix="") -> MetaVar:
        name = node.name + suffix
        if name not in meta_vars:
            meta_vars[name] = MetaVar(name, tuple(val.shape),
                                      _dtype_bytes(val.dtype))
        return meta_vars[name]

    for node in gm.graph.nodes:
        if node.op == "placeholder":
            val = node.meta.get("val")
            if isinstance(val, torch.Tensor):
                v = var_for(node, val)
                var_of[node.name] = v.name
                g.add_node(MetaNode(node.name, "placeholder", [], [v],
                                    is_placeholder=True))
            continue
        if node.op == "output":
            continue
        if node.op != "call_function":
            continue
        if node.target is operator.getitem:
            src, idx = node.args
            if isinstance(src, fx.Node) and f"{src.name}#{idx}" in meta_vars:
                var_of[node.name] = f"{src.name}#{idx}"
            continue
        val = node.meta.get("val")
        # collect tensor invars in pytree order (matches ShardAnnotation)
        invars: List[MetaVar] = []
        flat, _ = pytree.tree_flatten((node.args, node.kwargs))
        for a in flat:
            if isinstance(a, fx.Node):
                av = a.meta.get("val")
                if isinstance(av, torch.Tensor):
                    vn = var_of.get(a.name)
                    if vn is None:
                        continue
                    invars.append(meta_vars[vn])
        outvars: List[Optional[MetaVar]] = []
        if isinstance(val, torch.Tensor):
            v = var_for(node, val)
            var_of[node.name] = v.name
            outvars = [v]
        elif isinstance(val, (tuple, list)):
            for i, item in enumerate(val):
                if isinstance(item, torch.Tensor):
                    outvars.append(var_for(node, item, suffix=f"#{i}"))
                else:
                    outvars.append(None)
        else:
            continue   # non-tensor op: transform replicates it
        ann, combs = sharding_info.get(node.name, (None, {}))
        g.add_node(MetaNode(node.name, str(node.target), invars, outvars,
                            ann, combs, flops=_flops_of(node)))

    # outputs + constraints
    out_node = next(n for n in gm.graph.nodes if n.op == "output")
    out_args = out_node.args[0]
    output_constraints: Dict[str, object] = {}
    # invert state_io_map: src node name -> placeholder name
    state_out_of = {v: k for k, v in state_io_map.items() if v is not None}
    flat_outs, _ = pytree.tree_flatten(out_args)
    for o in flat_outs:
        if not isinstance(o, fx.Node):
            continue
        vn = var_of.get(o.name)
        if vn is None:
            continue
        g.output_vars.append(vn)
        if o.name in state_out_of:
            ph = state_out_of[o.name]
            if ph in var_of:
                g.state_io_map[var_of[ph]] = vn
                output_constraints[vn] = ("follow", var_of[ph])
        elif ret_names is not None and o.name in ret_names:
            # user-visible returns (loss etc.) come back replicated
            output_constraints[vn] = "replicate"
        # grads / untouched state: unconstrained, the solver decides
    return g, output_constraints, var_of
