"""Manual data-parallel family: ddp / zero2 / zero3 graph transforms.

Capability parity with reference ``easydist/torch/compile_dp.py``
(transform_ddp lines 55-80, transform_fsdp/zero 82-198, pre-scatter
320-333). Re-designed MI355X-first:

* the traced whole-step graph keeps ``aten._fused_adam_`` / ``_foreach_*``
  un-decomposed, so one graph node is one multi-tensor kernel launch at
  runtime (the reference decomposes then re-matches; we never decompose);
* ZeRO shards every tensor as a padded 1-D FLAT shard (``SPMD.FLAT``)
  instead of the reference's dim-0-divisibility special cases — uniform
  for any parameter shape, contiguous for RCCL reduce-scatter over xGMI;
* gradient reduction is reduce-scatter(avg)/all-reduce(avg) sized for the
  7-link xGMI ring: flat 1-D buffers, no per-dim chunk/cat relayout;
* zero3 gathers each parameter ONCE per step (288 GB HBM3E makes holding
  the gathered params through fwd+bwd the right trade; the reference
  re-gathers per use to fit smaller HBM).
"""
from __future__ import annotations

import logging
import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.fx as fx
import torch.utils._pytree as pytree

from ..metashard.metair import F, R, S, SPMD
from ..parallel.device_mesh import get_device_mesh
from ..runtime.compiled_func import EDCompiledFunc
from ..parallel import comm
from .tracing import ed_compile_func

logger = logging.getLogger(__name__)

aten = torch.ops.aten

FUSED_OPT_OPS = {
    aten._fused_adam_.default,
    aten._fused_adamw_.default,
    aten._fused_sgd_.default,
}


def _dp_group():
    mesh = get_device_mesh()
    if mesh is not None:
        return mesh.get_group(0)
    return dist.group.WORLD


# graph-visible collective wrappers (call_function targets) -------------------
def dp_all_reduce_avg(t: torch.Tensor) -> torch.Tensor:
    return comm.all_reduce(t, "avg", _dp_group())


def dp_reduce_scatter_flat(g: torch.Tensor) -> torch.Tensor:
    return comm.reduce_scatter_flat(g, "avg", _dp_group())


def dp_flat_shard(p: torch.Tensor) -> torch.Tensor:
    return comm.flat_shard_local(p, _dp_group())


def dp_gather_flat(shard: torch.Tensor, shape) -> torch.Tensor:
    return comm.all_gather_flat(shard, shape, _dp_group())


def _find_opt_node(graph: fx.Graph) -> Optional[fx.Node]:
    for n in graph.nodes:
        if n.op == "call_function" and n.target in FUSED_OPT_OPS:
            return n
    return None


def _compile_dp(func, mode: str, tracing_mode: str, args, kwargs, module,
                opt):
    """Trace + transform for parallel_mode in {ddp, zero2, zero3}."""
    assert mode in ("ddp", "zero2", "zero3")
    mesh = get_device_mesh()
    assert mesh is not None, "call set_device_mesh() before easydist_compile"
    world = mesh.size(0)
    device = "cuda" if torch.cuda.is_available() else "cpu"

    # decide data placements on the GLOBAL args, then trace at LOCAL shapes
    # (the traced graph bakes shapes into view/expand/backward ops, so the
    # graph must be built for the per-rank shard it will execute on)
    da_flat, da_spec = pytree.tree_flatten((args, kwargs))
    da_placements: List[Optional[List[SPMD]]] = []
    local_flat = []
    for v in da_flat:
        if isinstance(v, torch.Tensor) and v.ndim >= 1 \
                and v.shape[0] % world == 0 and world > 1:
            da_placements.append([S(0)])
            local_flat.append(
                comm.local_chunk(v, 0, mesh.get_group(0)))
        elif isinstance(v, torch.Tensor):
            da_placements.append([R])
            local_flat.append(v)
        else:
            da_placements.append(None)
            local_flat.append(v)
    largs, lkwargs = pytree.tree_unflatten(local_flat, da_spec)

    t0 = time.time()
    # keep optimizer ops whole: one node == one multi-tensor kernel launch
    params, buffers, named_states, gm = ed_compile_func(
        func, tracing_mode, largs, lkwargs, module, opt, decomp_table={})
    logger.info("[%s] traced %d nodes (%.2fs)", mode, len(gm.graph.nodes),
                time.time() - t0)

    graph = gm.graph
    placeholders = [n for n in graph.nodes if n.op == "placeholder"]
    pos_of = {n: i for i, n in enumerate(placeholders)}

    flat_inputs, in_spec = pytree.tree_flatten(
        (params, buffers, named_states, args, kwargs))
    n_params = len(params)
    n_state = (len(params) + len(buffers)
               + len(pytree.tree_flatten(named_states)[0]))

    out_node = next(n for n in graph.nodes if n.op == "output")
    flat_outs, out_spec_graph = pytree.tree_flatten(out_node.args[0])
    grad_slice = range(n_state, n_state + n_params)
    ret_slice = range(n_state + n_params, len(flat_outs))

    input_placements: List[Optional[List[SPMD]]] = [None] * len(flat_inputs)
    assert len(flat_inputs) == n_state + len(da_flat)
    for j, pl in enumerate(da_placements):
        input_placements[n_state + j] = pl
    for i in range(n_state):
        if isinstance(flat_inputs[i], torch.Tensor):
            input_placements[i] = [R]

    opt_node = _find_opt_node(graph)

    if mode == "ddp":
        _transform_ddp(graph, opt_node, flat_outs, grad_slice)
    else:
        assert opt_node is not None, (
            f"{mode} needs a fused optimizer (Adam/AdamW/SGD with "
            "fused=True); traced graph has no aten._fused_*_ node")
        _transform_zero(graph, gm, opt_node, flat_outs, grad_slice,
                        input_placements, pos_of, flat_inputs,
                        shard_param=(mode == "zero3"))

    # average user-visible returns (loss) so every rank sees the global value
    _avg_returns(graph, out_node, flat_outs, ret_slice)

    out_node.args = (pytree.tree_unflatten(flat_outs, out_spec_graph),)
    graph.lint()
    gm.recompile()

    # the make_fx pytree out-spec for (params, buffers, states, grads, ret)
    user_out_spec = getattr(gm, "_out_spec", None)

    # strip pytree codegen: runtime passes the flat placeholder list
    gm.graph._codegen = fx.graph.CodeGen()
    gm.recompile()

    # io map: state input position -> output position holding its new value
    io_pos_map: Dict[int, int] = {}
    name_to_out_pos = {}
    for k, o in enumerate(flat_outs):
        if isinstance(o, fx.Node):
            name_to_out_pos.setdefault(o.name, k)
    # placeholders that appear directly in outputs (in-place updated state)
    for ph, i in pos_of.items():
        if i < n_state and ph.name in name_to_out_pos:
            io_pos_map[i] = name_to_out_pos[ph.name]
    # zero2 params: output is the gathered full tensor node
    for i, o in enumerate(flat_outs[:n_state]):
        if isinstance(o, fx.Node) and o.op != "placeholder":
            io_pos_map[i] = name_to_out_pos[o.name]

    output_placements = [None] * len(flat_outs)

    compiled = EDCompiledFunc(
        gm, in_spec, out_spec_graph, input_placements, output_placements,
        [i for i in range(n_state)
         if isinstance(flat_inputs[i], torch.Tensor)],
        io_pos_map, n_params, list(params.keys()), device)
    compiled.init_named_states = named_states
    compiled.meta = {"search_time": 0.0, "solve_time": 0.0,
                     "n_nodes": len(gm.graph.nodes),
                     "out_spec": user_out_spec, "parallel_mode": mode}
    return compiled


def _transform_ddp(graph: fx.Graph, opt_node, flat_outs, grad_slice):
    """all_reduce(avg) every grad before any consumer (optimizer + output).

    reference: compile_dp.py:55-80. The grads feeding the optimizer are the
    fused-op node's second list argument (the grad OUTPUTS of the traced
    step are None whenever the user calls opt.zero_grad at step end, so
    they cannot be used to locate the live grad nodes)."""
    done = {}

    def reduced(g):
        if g not in done:
            with graph.inserting_after(g):
                done[g] = graph.call_function(dp_all_reduce_avg, (g,))
        return done[g]

    if opt_node is not None:
        fused_args = list(opt_node.args)
        fused_args[1] = [reduced(g) for g in fused_args[1]]
        opt_node.args = tuple(fused_args)
    for i in grad_slice:
        g = flat_outs[i]
        if isinstance(g, fx.Node):
            flat_outs[i] = reduced(g)
    assert done, ("ddp: no gradient nodes found — use a fused optimizer "
                  "(fused=True) or return grads from the train step")


def _transform_zero(graph: fx.Graph, gm, opt_node: fx.Node, flat_outs,
                    grad_slice, input_placements, pos_of, flat_inputs,
                    shard_param: bool):
    """ZeRO-2/3 around the fused optimizer node.

    reference: compile_dp.py:82-198 (transform_fsdp). Optimizer states are
    FLAT-sharded persistent inputs; grads reduce-scattered to flat shards;
    params are flat-sharded in the graph (zero2) or persistently (zero3)
    and the updated full params rebuilt with one all-gather per param.
    """
    fused_args = list(opt_node.args)
    param_nodes: List[fx.Node] = list(fused_args[0])
    grad_nodes: List[fx.Node] = list(fused_args[1])
    state_lists = [list(l) for l in fused_args[2:] if isinstance(l, (list, tuple))]

    # state tensor lists: everything after grads except state_steps scalars
    for lst in state_lists:
        for s in lst:
            if not isinstance(s, fx.Node) or s.op != "placeholder":
                continue
            i = pos_of[s]
            v = flat_inputs[i]
            if isinstance(v, torch.Tensor) and v.numel() > 1:
                input_placements[i] = [F(v.shape)]

    # grads -> flat reduce-scatter(avg) shards
    new_grads = []
    for g in grad_nodes:
        with graph.inserting_after(g):
            rs = graph.call_function(dp_reduce_scatter_flat, (g,))
        new_grads.append(rs)
    fused_args[1] = new_grads

    # user-returned grads: without this, the step's grad outputs are full
    # unreduced per-rank partials (rank-divergent, unlike the ddp path's
    # averaged grads). Rebuild the averaged full grad from the
    # reduce-scattered shard with one all-gather per RETURNED grad only.
    rs_of = dict(zip(grad_nodes, new_grads))
    regathered: Dict[fx.Node, fx.Node] = {}
    for i in grad_slice:
        g = flat_outs[i]
        if not (isinstance(g, fx.Node) and g in rs_of):
            continue
        if g not in regathered:
            val = g.meta.get("val") if hasattr(g, "meta") else None
            if not isinstance(val, torch.Tensor):
                continue
            with graph.inserting_after(rs_of[g]):
                regathered[g] = graph.call_function(
                    dp_gather_flat, (rs_of[g], tuple(val.shape)))
        flat_outs[i] = regathered[g]

    new_params = []
    gathered: Dict[fx.Node, fx.Node] = {}
    for p in param_nodes:
        i = pos_of[p]
        shape = tuple(flat_inputs[i].shape)
        if shard_param:
            # zero3: placeholder holds the persistent flat shard; gather the
            # full param once, up front, for every fwd/bwd use
            input_placements[i] = [F(shape)]
            with graph.inserting_after(p):
                full = graph.call_function(dp_gather_flat, (p, shape))
            p.replace_all_uses_with(
                full, delete_user_cb=lambda u: u is not full
                and u is not opt_node and u.op != "output")
            new_params.append(p)
        else:
            # zero2: param enters full; optimizer sees my flat shard
            with graph.inserting_before(opt_node):
                sh = graph.call_function(dp_flat_shard, (p,))
            new_params.append(sh)
            gathered[p] = sh
    fused_args[0] = new_params
    opt_node.args = tuple(fused_args)

    if not shard_param:
        # zero2: rebuild full updated params after the (in-place) step
        last = opt_node
        for p in param_nodes:
            i = pos_of[p]
            shape = tuple(flat_inputs[i].shape)
            with graph.inserting_after(last):
                full = graph.call_function(dp_gather_flat,
                                           (gathered[p], shape))
            last = full
            # param state output -> gathered updated param
            for k in range(len(flat_outs)):
                if flat_outs[k] is p:
                    flat_outs[k] = full


def _avg_returns(graph: fx.Graph, out_node: fx.Node, flat_outs, ret_slice):
    """Average scalar floating returns (the loss) so every rank sees the
    global-batch value. Non-scalar returns stay local: without placement
    tracking (that is the auto path's job) a batch-sharded return cannot be
    safely reassembled here."""
    done = {}
    for i in ret_slice:
        r = flat_outs[i]
        if not isinstance(r, fx.Node):
            continue
        if r in done:
            flat_outs[i] = done[r]
            continue
        val = r.meta.get("val") if hasattr(r, "meta") else None
        if not (isinstance(val, torch.Tensor) and val.ndim == 0
                and val.is_floating_point()):
            continue
        with graph.inserting_before(out_node):
            ar = graph.call_function(dp_all_reduce_avg, (r,))
        done[r] = ar
        flat_outs[i] = ar
