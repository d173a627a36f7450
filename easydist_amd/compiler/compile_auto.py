"""The auto-SPMD compile pipeline.

Capability parity with reference ``easydist/torch/compile_auto.py``
(_compile_auto, lines 456-822): trace -> canonicalize -> rule discovery ->
per-mesh-dim solve -> sharding transform -> runtime.

Differences by design (MI355X-first):
* every rank runs the (deterministic) discovery+solve instead of rank-0 +
  torch-RPC broadcast — the MILP is deterministic, ranks agree without a
  control-plane exchange; a broadcast fallback verifies agreement;
* the traced graph is made fully pure (no copy_) before transforming, so
  hipGraph capture and static memory planning see a functional program.
"""
from __future__ import annotations

import logging
import time
from typing import Dict, List, Optional

import torch
import torch.utils._pytree as pytree

from .. import config as mdconfig
from ..autoflow.solver import AutoFlowSolver1D
from ..metashard.metair import R, SPMD
from ..parallel.device_mesh import get_device_mesh
from ..runtime.compiled_func import EDCompiledFunc
from .bridge import fx2meta_graph
from .passes.functionalize import canonicalize
from .passes.sharding import sharding_transform
from .sharding_interpreter import EDTorchShardingAnn
from .tracing import ed_compile_func

logger = logging.getLogger(__name__)


def _compile_auto(func, tracing_mode, args, kwargs, module, opt):
    mesh = get_device_mesh()
    assert mesh is not None, "call set_device_mesh() before easydist_compile"
    device = ("cuda" if torch.cuda.is_available() else "cpu")

    # ---- 1. trace --------------------------------------------------------
    t0 = time.time()
    params, buffers, named_states, gm = ed_compile_func(
        func, tracing_mode, args, kwargs, module, opt)
    logger.info("traced whole-step graph: %d nodes (%.2fs)",
                len(gm.graph.nodes), time.time() - t0)

    # ---- 2. canonicalize -------------------------------------------------
    gm, io_map = canonicalize(gm)

    # flat input bookkeeping (must match make_fx's pytree flattening)
    flat_inputs, in_spec = pytree.tree_flatten(
        (params, buffers, named_states, args, kwargs))
    placeholders = [n for n in gm.graph.nodes if n.op == "placeholder"]
    n_state = (len(params) + len(buffers)
               + len(pytree.tree_flatten(named_states)[0]))
    state_positions = [i for i in range(min(n_state, len(placeholders)))
                       if isinstance(flat_inputs[i], torch.Tensor)]

    out_node = next(n for n in gm.graph.nodes if n.op == "output")
    flat_outs, out_spec_graph = pytree.tree_flatten(out_node.args[0])
    # names of user-visible return nodes: everything past states+grads
    n_state_outs = n_state
    n_grads = len(params)
    ret_names = set()
    for o in flat_outs[n_state_outs + n_grads:]:
        if hasattr(o, "name"):
            ret_names.add(o.name)

    gm, out_pl_env, search_time, solve_time = shard_graph(
        gm, mesh, io_map, ret_names, device, n_state=n_state)

    # ---- 5b. comm dedup + overlap ----------------------------------------
    from .passes.comm_optimize import comm_cse, comm_optimize
    comm_cse(gm)
    comm_optimize(gm)

    # ---- 5b2. lower hot aten ops to the gfx950 kernels -------------------
    from .passes.lower_hip import (lower_attn_pack, lower_bias_grad_fuse,
                                   lower_cross_entropy, lower_gelu_bwd_fuse,
                                   lower_gelu_fwd_fuse, lower_gemm,
                                   lower_layer_norm, lower_sdpa)
    import os as _os
    lower_layer_norm(gm)
    lower_sdpa(gm)
    if _os.environ.get("EASYDIST_ATTN_PACK", "1") != "0":
        lower_attn_pack(gm)
    lower_cross_entropy(gm)
    if mdconfig.hip_gemm:
        lower_gemm(gm)
        lower_gelu_bwd_fuse(gm)
        if _os.environ.get("EASYDIST_GELU_FWD_FUSE", "1") != "0":
            lower_gelu_fwd_fuse(gm)
        lower_bias_grad_fuse(gm)

    # ---- 5c. re-fuse the decomposed Adam chains into ONE kernel ----------
    if opt is not None and getattr(mdconfig, "fuse_optimizer", True):
        from .passes.fuse_optimizer import fuse_optimizer
        out_node_f = next(n for n in gm.graph.nodes if n.op == "output")
        flat_outs_f, spec_f = pytree.tree_flatten(out_node_f.args[0])
        if type(opt).__name__ == "SGD":
            ppos = _sgd_positions(params, buffers, named_states)
        else:
            ppos = _adam_positions(params, buffers, named_states)
        if ppos:
            nfused = fuse_optimizer(gm, flat_outs_f,
                                    [n for n in gm.graph.nodes
                                     if n.op == "placeholder"], ppos, opt,
                                    pl_env=out_pl_env)
            if nfused:
                out_node_f.args = (pytree.tree_unflatten(flat_outs_f,
                                                         spec_f),)
                gm.graph.eliminate_dead_code()
                gm.graph.lint()
                gm.recompile()

    from ..utils.dumps import dump_graph, dump_graph_dot
    dump_graph(gm, "auto_sharded")
    dump_graph_dot(gm, "auto_sharded")

    # strip the pytree codegen: the runtime calls the graph with the flat
    # placeholder list and receives the flat output list
    import torch.fx as _fx
    gm.graph._codegen = _fx.graph.CodeGen()
    gm.recompile()

    # ---- 6. runtime ------------------------------------------------------
    input_placements = []
    for i, ph in enumerate(placeholders):
        pls = out_pl_env.get(ph.name)
        input_placements.append(pls[0] if pls else None)
    # pad for non-placeholder flat inputs (shouldn't happen, but be safe)
    while len(input_placements) < len(flat_inputs):
        input_placements.append(None)

    # flat-position io map: placeholder idx -> output idx
    name_to_out_pos = {}
    for k, o in enumerate(flat_outs):
        if hasattr(o, "name"):
            name_to_out_pos.setdefault(o.name, k)
    io_pos_map = {}
    ph_by_name = {ph.name: i for i, ph in enumerate(placeholders)}
    for ph_name, src_name in io_map.items():
        if ph_name in ph_by_name and src_name in name_to_out_pos:
            io_pos_map[ph_by_name[ph_name]] = name_to_out_pos[src_name]
    # positional fallback: non-fused optimizers trace as in-place ops
    # (empty io_map after fix_inplace); output i of the state block IS the
    # new value of placeholder i
    for i in state_positions:
        if i not in io_pos_map and i < len(flat_outs):
            o = flat_outs[i]
            if hasattr(o, "name") and (i >= len(placeholders)
                                       or o.name != placeholders[i].name):
                io_pos_map[i] = i

    output_placements = []
    for o in flat_outs:
        if hasattr(o, "name") and o.name in out_pl_env:
            output_placements.append(out_pl_env[o.name][0])
        else:
            output_placements.append(None)

    compiled = EDCompiledFunc(
        gm, in_spec, out_spec_graph, input_placements, output_placements,
        state_positions, io_pos_map, len(params), list(params.keys()), device)
    compiled.init_named_states = named_states
    compiled.ret_out_positions = list(range(n_state_outs + n_grads,
                                            len(flat_outs)))
    # qualified names for checkpoint APIs: position -> user-facing name
    qualnames = (list(params.keys()) + list(buffers.keys())
                 + [f"{pn}.{k}" for pn, st in named_states.items()
                    for k, v in st.items()
                    if isinstance(v, torch.Tensor)])
    compiled.state_qualnames = {i: qn for i, qn in enumerate(qualnames)}
    compiled.debug_pl_env = out_pl_env   # node name -> chosen placements
    compiled.meta = {
        "search_time": search_time, "solve_time": solve_time,
        "n_nodes": len(gm.graph.nodes), "out_spec": gm._out_spec
        if hasattr(gm, "_out_spec") else None,
    }
    return compiled


def shard_graph(gm, mesh, io_map, ret_names, device, fix_rets=True,
                n_state=None):
    """Discovery + per-mesh-dim MILP solve + sharding transform.

    Shards along the mesh's spmd dims only (a 'pp' dim is skipped), so
    the hybrid pipeline path can run this on a marker-carrying graph
    before splitting stages. Returns (gm, out_pl_env, search_s, solve_s).
    """
    cache_path = _strategy_cache_path(gm, mesh)
    cached = _load_strategy_cache(cache_path)
    if cached is not None:
        strategies_per_dim, search_time, solve_time = cached
        logger.info("strategy cache hit: %s", cache_path)
    else:
        t0 = time.time()
        sharding_info = EDTorchShardingAnn(
            gm, device=mdconfig.discovery_device or device).run()
        search_time = time.time() - t0
        logger.info("sharding discovery: %d annotated ops (%.2fs)",
                    len(sharding_info), search_time)

        meta_graph, output_constraints, var_of = fx2meta_graph(
            gm, sharding_info, io_map, ret_names)
        spmd_dims = mesh.spmd_dims()
        strategies_per_dim = []
        already_sharded: Dict[str, Dict[int, int]] = {}
        t0 = time.time()
        for mesh_dim in range(mesh.ndim):
            size = mesh.size(mesh_dim)
            if mesh_dim not in spmd_dims or size == 1:
                strategies_per_dim.append({})
                continue
            clusters = meta_graph.coarsen(
                mdconfig.coarsen_level if mdconfig.enable_graph_coarsen
                else 0)
            solver = AutoFlowSolver1D(meta_graph, size, already_sharded,
                                      output_constraints)
            solver.add_coarsen_graph(clusters)
            use_beam = (mdconfig.solver_mode == "beam"
                        or len(clusters) > mdconfig.ilp_max_clusters)
            if use_beam and mdconfig.solver_mode != "beam":
                logger.info("%d clusters > ilp_max_clusters=%d: using beam "
                            "search (the timed-out MILP incumbent is worse "
                            "than the beam solution at this scale)",
                            len(clusters), mdconfig.ilp_max_clusters)
            choice = (solver.beam_search() if use_beam
                      else solver.ilp_solve())
            node_strats: Dict = {}
            for st in choice.values():
                node_strats.update(st.node_strategies)
            strategies_per_dim.append(node_strats)
            # update already_sharded for the next dim
            for st in choice.values():
                for v, pl in st.out_placements.items():
                    if pl.is_shard():
                        already_sharded.setdefault(v, {})
                        already_sharded[v][pl.dim] = \
                            already_sharded[v].get(pl.dim, 1) * size
        solve_time = time.time() - t0
        logger.info("strategy solve: %.2fs (search %.2fs)", solve_time,
                    search_time)
        _save_strategy_cache(cache_path,
                             (strategies_per_dim, search_time, solve_time))

    _verify_strategy_agreement(strategies_per_dim)

    from ..utils.dumps import dump_graph, dump_strategies
    dump_strategies(strategies_per_dim, "auto")
    dump_graph(gm, "auto_pre_shard")

    gm, out_pl_env = sharding_transform(gm, strategies_per_dim, mesh.shape)
    gm = _fix_output_reshard(gm, out_pl_env, io_map,
                             ret_names if fix_rets else set(), mesh,
                             n_state=n_state)
    return gm, out_pl_env, search_time, solve_time


def _strategy_cache_path(gm, mesh) -> Optional[str]:
    """Strategy cache keyed by (graph text, mesh shape). reference:
    compile_auto.py:97-106 (pickle keyed by input signature)."""
    import hashlib
    import os
    if not mdconfig.enable_compile_cache:
        return None
    h = hashlib.sha256()
    h.update(str(gm.graph).encode())
    h.update(repr(mesh.shape).encode())
    import easydist_amd
    h.update(easydist_amd.__version__.encode())
    h.update(torch.__version__.encode())
    # rule changes must invalidate cached strategies
    from .preset_propagation import _PRESET_REGISTRY
    h.update(repr(sorted(str(k) for k in _PRESET_REGISTRY)).encode())
    h.update(f"{mdconfig.solver_mode}/{mdconfig.beam_width}/"
             f"{mdconfig.ilp_max_clusters}".encode())
    d = os.path.join(os.path.expanduser("~"), ".easydist_amd",
                     "compile_cache")
    os.makedirs(d, exist_ok=True)
    return os.path.join(d, h.hexdigest()[:24] + ".pkl")


def _load_strategy_cache(path):
    import os
    import pickle
    if path is None or not os.path.exists(path):
        return None
    try:
        with open(path, "rb") as f:
            return pickle.load(f)
    except Exception:
        return None


def _save_strategy_cache(path, payload):
    import os
    import pickle
    if path is None:
        return
    try:
        tmp = f"{path}.{os.getpid()}.tmp"
        with open(tmp, "wb") as f:
            pickle.dump(payload, f)
        os.replace(tmp, path)   # atomic: concurrent ranks never read partial
    except Exception as e:   # noqa: BLE001
        logger.warning("strategy cache write failed: %s", e)


def _verify_strategy_agreement(strategies_per_dim):
    """All ranks must have solved to the SAME strategy (the MILP is
    deterministic, but a divergence would deadlock the collectives —
    fail loudly instead). reference control plane: compile_auto.py:517
    (rank0 + RPC broadcast); here every rank solves and we cross-check a
    hash."""
    import hashlib

    import torch.distributed as dist
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return
    digest = hashlib.sha256(
        repr([sorted((k, repr(v)) for k, v in d.items())
              for d in strategies_per_dim]).encode()).digest()[:8]
    t = torch.tensor(list(digest), dtype=torch.uint8)
    if torch.cuda.is_available():
        t = t.cuda()
    ref = t.clone()
    dist.broadcast(ref, src=0)
    mismatch = torch.tensor(
        [0 if torch.equal(t, ref) else 1],
        device=t.device)
    dist.all_reduce(mismatch, op=dist.ReduceOp.MAX)
    if int(mismatch):
        # resolve by adopting rank 0's strategies (mirrors the reference's
        # rank0-solves + broadcast control plane); ALL ranks join the
        # object broadcast
        if not torch.equal(t, ref):
            logger.warning("strategy solve diverged from rank 0: adopting "
                           "rank 0's result")
        # copy first: on rank 0 payload[0] IS strategies_per_dim and
        # clear()+extend(itself) would leave rank 0 with an empty list
        payload = [list(strategies_per_dim)]
        dist.broadcast_object_list(payload, src=0)
        if not torch.equal(t, ref):
            strategies_per_dim.clear()
            strategies_per_dim.extend(payload[0])


def _adam_positions(params, buffers, named_states):
    """Map each Adam-stepped param to its flat input/output positions.

    Flat layout (pytree insertion order): params..., buffers...,
    named_states[pname][state_key]... — identical on the input and output
    side (the traced step returns the same structure)."""
    param_names = list(params.keys())
    n_p, n_b = len(params), len(buffers)
    offset = n_p + n_b
    out = {}
    for pname, st in named_states.items():
        if pname not in param_names:
            offset += len([v for v in st.values()
                           if isinstance(v, torch.Tensor)])
            continue
        p_pos = param_names.index(pname)
        entry = {"param": p_pos}
        for key, val in st.items():
            if not isinstance(val, torch.Tensor):
                continue
            if key == "step":
                entry["step"] = offset
                entry["step_in"] = offset
            elif key == "exp_avg":
                entry["exp_avg"] = offset
                entry["exp_avg_in"] = offset
            elif key == "exp_avg_sq":
                entry["exp_avg_sq"] = offset
                entry["exp_avg_sq_in"] = offset
            offset += 1
        if {"step", "exp_avg", "exp_avg_sq"} <= set(entry):
            out[p_pos] = entry
    return out


def _sgd_positions(params, buffers, named_states):
    """Map each momentum-SGD param to its flat positions (momentum_buffer
    state key); same flat-layout contract as _adam_positions."""
    param_names = list(params.keys())
    offset = len(params) + len(buffers)
    out = {}
    for pname, st in named_states.items():
        n_t = len([v for v in st.values() if isinstance(v, torch.Tensor)])
        if pname not in param_names or "momentum_buffer" not in st \
                or not isinstance(st.get("momentum_buffer"), torch.Tensor):
            offset += n_t
            continue
        p_pos = param_names.index(pname)
        entry = {"param": p_pos}
        for key, val in st.items():
            if not isinstance(val, torch.Tensor):
                continue
            if key == "momentum_buffer":
                entry["buf"] = offset
                entry["buf_in"] = offset
            offset += 1
        out[p_pos] = entry
    return out


def _fix_output_reshard(gm, out_pl_env, io_map, ret_names, mesh,
                        n_state=None):
    """Reshard outputs that must land at a fixed placement: user returns to
    REPLICATE; state outputs back to their input placeholder's placement
    (reference behavior: sharding.py:920-949). State round-trips resolve
    through io_map when the trace used copy_ AND positionally (output i of
    the state block is placeholder i) for in-place-op traces whose io_map
    is empty after fix_inplace."""
    from ..runtime import comm_runtime as crt
    from .passes.sharding import ShardingTransform
    graph = gm.graph
    out_node = next(n for n in graph.nodes if n.op == "output")
    flat_outs, spec = pytree.tree_flatten(out_node.args[0])
    ph_pl = {}
    for n in graph.nodes:
        if n.op == "placeholder" and n.name in out_pl_env:
            ph_pl[n.name] = out_pl_env[n.name][0]
    src_to_ph = {v: k for k, v in io_map.items() if v is not None}

    placeholders = [n for n in graph.nodes if n.op == "placeholder"]
    tr = ShardingTransform(gm, [], mesh.shape)
    tr.out_pl = out_pl_env
    changed = False
    with graph.inserting_before(out_node):
        for i, o in enumerate(flat_outs):
            if not hasattr(o, "name") or o.name not in out_pl_env:
                continue
            cur = out_pl_env[o.name][0]
            want = None
            if o.name in ret_names:
                want = [R] * mesh.ndim
            elif o.name in src_to_ph and src_to_ph[o.name] in ph_pl:
                want = ph_pl[src_to_ph[o.name]]
            elif (n_state is not None and i < n_state
                  and i < len(placeholders)
                  and o.name != placeholders[i].name):
                want = ph_pl.get(placeholders[i].name, [R] * mesh.ndim)
            if want is None:
                continue
            if all(repr(c) == repr(w) for c, w in zip(cur, want)):
                continue
            new = tr._reshard(graph, o, cur, want)
            out_pl_env[new.name] = [want]
            flat_outs[i] = new
            changed = True
    if changed:
        out_node.args = (pytree.tree_unflatten(flat_outs, spec),)
        graph.lint()
        gm.recompile()
    return gm
