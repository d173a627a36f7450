"""Pipeline compile entry. reference: easydist/torch/experimental/pp/api.py.

``easydist_compile(parallel_mode='pp', nstages=4, nchunks=8,
schedule='dapple')(train_step)`` — or pass ``split_points={'h.5'}`` for
explicit boundaries. With torch.distributed initialized and world_size ==
nstages each rank runs its stage; otherwise a single-process local runtime
executes all stages (the reference's `local_pp_stage_cnt` mode,
pp/api.py:33-82).
"""
from __future__ import annotations

import logging
from typing import Optional, Set

import torch
import torch.distributed as dist
import torch.utils._pytree as pytree

from ...compiler.tracing import ed_compile_func
from .compile_pipeline import compile_pipeline
from .runtime import LocalPipelineRuntime, PipelineStage, _chunk
from .split import SplitPatcher, annotate_split_points, split_into_equal_size

logger = logging.getLogger(__name__)


def _compile_pp(func, tracing_mode, args, kwargs, module, opt,
                nstages: Optional[int] = None,
                split_points: Optional[Set[str]] = None,
                nchunks: int = 4, schedule: str = "gpipe",
                scale_grads: bool = True):
    assert module is not None, "pp needs the nn.Module argument"
    if split_points:
        annotate_split_points(module, set(split_points))
        if nstages is None:
            nstages = len(split_points) + 1
    else:
        assert nstages is not None, "pass nstages= or split_points="
        split_into_equal_size(nstages)(module)

    device = "cuda" if torch.cuda.is_available() else "cpu"

    # trace at MICROBATCH shapes: chunk the data args, trace on chunk 0
    # (reference: pp/api.py traces on chunk 0 of the microbatched inputs)
    da_flat, da_spec = pytree.tree_flatten((args, kwargs))
    chunk0 = [(_chunk(v, nchunks)[0]
               if isinstance(v, torch.Tensor) and v.ndim >= 1 else v)
              for v in da_flat]
    cargs, ckwargs = pytree.tree_unflatten(chunk0, da_spec)

    patcher = SplitPatcher(module, opt)
    params, buffers, named_states, gm = ed_compile_func(
        func, tracing_mode, cargs, ckwargs, module, opt,
        split_patcher_ctx=patcher)
    logger.info("[pp] traced %d nodes", len(gm.graph.nodes))

    flat_inputs, _ = pytree.tree_flatten(
        (params, buffers, named_states, cargs, ckwargs))
    n_params = len(params)
    n_state = (len(params) + len(buffers)
               + len(pytree.tree_flatten(named_states)[0]))

    # ---- hybrid pp x spmd: auto-SPMD-shard the marker-carrying graph
    # along the spmd mesh dims BEFORE stage splitting
    # (reference: compile_auto.py:683-715 + test_hybrid.py meshes)
    from ...parallel.device_mesh import get_device_mesh
    mesh = get_device_mesh()
    hybrid = (mesh is not None and mesh.has_dim("pp")
              and mesh.size() > mesh.size(mesh.dim_index("pp")))
    io_map = None
    out_pl_env = {}
    if hybrid:
        from ...compiler.compile_auto import shard_graph
        from ...compiler.passes.functionalize import canonicalize
        gm, io_map = canonicalize(gm)
        gm, out_pl_env, _st, _sv = shard_graph(gm, mesh, io_map, set(),
                                               device, fix_rets=False,
                                               n_state=n_state)

    info = compile_pipeline(gm, flat_inputs, n_params, n_state,
                            list(params.keys()), io_map=io_map)
    assert info.nstages == nstages, \
        f"trace produced {info.nstages} stages, expected {nstages}"

    if hybrid:
        info.pp_mesh_dim = mesh.dim_index("pp")
        for name in info.ph_names:
            pls = out_pl_env.get(name)
            if pls:
                info.ph_placements[name] = pls[0]
        import os as _os
        if _os.environ.get("EASYDIST_DEBUG_PP"):
            logger.error("ph_placements sample: %s",
                         {k: repr(v) for k, v in
                          list(info.ph_placements.items())[:6]})
        # ret + boundary placements for the runtime's spmd fixes
        for sg in info.stages:
            for name in sg.ret_names:
                pls = out_pl_env.get(name)
                if pls:
                    info.ret_placements[name] = pls[0]
            for name in sg.fw_recv + sg.bw_recv:
                pls = out_pl_env.get(name)
                if pls:
                    info.boundary_placements[name] = pls[0]

    qualnames = (list(params.keys()) + list(buffers.keys())
                 + [f"{pn}.{k}" for pn, st in named_states.items()
                    for k, v in st.items()
                    if isinstance(v, __import__("torch").Tensor)])
    for ph, qn in zip(info.ph_names, qualnames):
        info.ph_qualnames[ph] = qn

    ph_values = dict(zip(info.ph_names, flat_inputs))

    world = dist.get_world_size() if dist.is_initialized() else 1
    if hybrid and world > 1:
        assert world == mesh.size(), \
            f"world {world} != mesh size {mesh.size()}"
        assert mesh.size(info.pp_mesh_dim) == nstages
        stage_idx = mesh.my_coords()[info.pp_mesh_dim]
        rt = PipelineStage(info, stage_idx, device, nchunks,
                           schedule=schedule, scale_grads=scale_grads)
    elif world > 1:
        assert world == nstages, \
            f"world size {world} must equal nstages {nstages}"
        rt = PipelineStage(info, dist.get_rank(), device, nchunks,
                           schedule=schedule, scale_grads=scale_grads)
    else:
        rt = LocalPipelineRuntime(info, device, nchunks,
                                  scale_grads=scale_grads)
    rt.init_state(ph_values)
    return rt
