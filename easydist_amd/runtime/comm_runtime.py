"""Graph-callable communication targets.

These are the `call_function` targets the sharding transform inserts. Each
resolves its process group from the global device mesh at call time, so the
transformed graph is picklable and mesh-relative. The start/wait split keeps
communication graph-visible for the overlap passes (RCPSP, tile_comm), and
RCCL launches the collective on its own HIP stream under the hood.
"""
from __future__ import annotations

import torch

from ..parallel import comm
from ..parallel.device_mesh import get_device_mesh


def _group(mesh_dim: int):
    mesh = get_device_mesh()
    assert mesh is not None, "set_device_mesh() before running a sharded graph"
    return mesh.get_group(mesh_dim)


def rt_all_reduce_start(t, op: str, mesh_dim: int):
    return comm.all_reduce_start(t, op, _group(mesh_dim))


def rt_all_gather_start(t, gather_dim: int, mesh_dim: int):
    return comm.all_gather_start(t, gather_dim, _group(mesh_dim))


def rt_reduce_scatter_start(t, scatter_dim: int, op: str, mesh_dim: int):
    return comm.reduce_scatter_start(t, scatter_dim, op, _group(mesh_dim))


def rt_all_to_all_start(t, src_dim: int, dst_dim: int, mesh_dim: int):
    return comm.all_to_all_start(t, src_dim, dst_dim, _group(mesh_dim))


def rt_wait(w):
    return comm.comm_wait(w)


def rt_local_chunk(t, dim: int, mesh_dim: int):
    return comm.local_chunk(t, dim, _group(mesh_dim))


def rt_partial_localize(t, mesh_dim: int):
    return comm.partial_localize(t, _group(mesh_dim))


# names used by graph printing / the comm-optimize pass
COMM_START_TARGETS = {rt_all_reduce_start, rt_all_gather_start,
                      rt_reduce_scatter_start, rt_all_to_all_start}
COMM_LOCAL_TARGETS = {rt_local_chunk, rt_partial_localize}


def rt_ring_attention(q, k, v, causal: bool, mesh_dim: int):
    """Sequence-parallel exact attention over the xGMI ring (SP chosen by
    the solver: flash_attention with S(seq) inputs is rewritten to this —
    passes/sharding.py). Returns (out_local, lse_local)."""
    from ..ops import ring_attention as ra
    g = _group(mesh_dim)
    import torch.distributed as dist
    if g is None or dist.get_world_size(g) == 1:
        return torch.ops.easydist_amd.flash_attention(q, k, v, causal)
    out, lse = ra._ring_forward(q, k, v, g, causal)
    return out, lse


def rt_ring_attention_bwd(grad, q, k, v, out, lse, causal: bool,
                          mesh_dim: int):
    from ..ops import ring_attention as ra
    g = _group(mesh_dim)
    import torch.distributed as dist
    if g is None or dist.get_world_size(g) == 1:
        return torch.ops.easydist_amd.flash_attention_bwd(
            grad, q, k, v, out, lse, causal)
    return ra.ring_bwd(grad, q, k, v, out, lse, g, causal)


def rt_grouped_all_reduce_start(tensors, op: str, mesh_dim: int):
    return comm.all_reduce_bucket_start(list(tensors), op, _group(mesh_dim))


def rt_grouped_all_gather_start(tensors, gather_dims, mesh_dim: int):
    return comm.all_gather_bucket_start(list(tensors), list(gather_dims),
                                        _group(mesh_dim))


def rt_grouped_wait(w):
    return comm.comm_wait(w)


COMM_START_TARGETS.add(rt_grouped_all_reduce_start)
COMM_START_TARGETS.add(rt_grouped_all_gather_start)


def rt_p2p_reshard(t, global_shape, cur, want):
    """Rectangle-intersection P2P reshard over the whole mesh (emitted by
    the sharding transform for multi-mesh-dim S->S transitions)."""
    mesh = get_device_mesh()
    return comm.p2p_reshard(t, list(global_shape), [tuple(p) for p in cur],
                            [tuple(p) for p in want], mesh)
