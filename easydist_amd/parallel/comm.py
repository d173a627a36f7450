"""Collective wrappers the sharded graph calls — RCCL over xGMI.

The inventory matches the reference's collective call sites
(easydist/torch/passes/sharding.py:94-163): all_reduce, all_gather along an
arbitrary dim, reduce_scatter along an arbitrary dim, all_to_all, local
chunk/scatter. Differences by design:

* every wrapper is split into an async ``*_start`` returning a work handle
  and a ``comm_wait`` — kept graph-visible so overlap passes (RCPSP /
  tile_comm) can reorder them;
* ``all_to_all`` is a REAL single-shot `all_to_all_single` (pairwise xGMI
  exchange under RCCL) — the reference faked it as all-gather + slice
  (sharding.py:155-163 TODO);
* on a `gloo` backend (CPU tests) the collectives gloo lacks
  (reduce_scatter, all_to_all) are emulated, so the multi-process CPU test
  suite exercises the same graph code paths the MI355X runs.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

REDUCE_OPS = {
    "sum": dist.ReduceOp.SUM,
    "avg": dist.ReduceOp.AVG,
    "max": dist.ReduceOp.MAX,
    "min": dist.ReduceOp.MIN,
}


def _is_gloo(group) -> bool:
    try:
        return dist.get_backend(group) == "gloo"
    except Exception:
        return False


def _rank(group):
    return dist.get_rank(group)


def _world(group):
    return dist.get_world_size(group)


class _Work:
    """Pending communication: result tensor + optional dist work handle."""
    __slots__ = ("tensor", "work", "post")

    def __init__(self, tensor, work=None, post=None):
        self.tensor = tensor
        self.work = work
        self.post = post


def comm_wait(w):
    if isinstance(w, _Work):
        if w.work is not None:
            w.work.wait()
        t = w.tensor
        if w.post is not None:
            t = w.post(t)
        return t
    return w


# ----------------------------------------------------------- all_reduce ------
def all_reduce_start(t: torch.Tensor, op: str = "sum", group=None) -> _Work:
    t = t.contiguous()
    red = REDUCE_OPS[op]
    if op == "avg" and _is_gloo(group):
        red = dist.ReduceOp.SUM
        n = _world(group)
        work = dist.all_reduce(t, op=red, group=group, async_op=True)
        return _Work(t, work, post=lambda x: x / n)
    work = dist.all_reduce(t, op=red, group=group, async_op=True)
    return _Work(t, work)


def all_reduce(t, op: str = "sum", group=None):
    return comm_wait(all_reduce_start(t, op, group))


# ----------------------------------------------------------- all_gather ------
def all_gather_start(t: torch.Tensor, gather_dim: int = 0, group=None) -> _Work:
    """SHARD(dim) -> REPLICATE. Gathers along dim 0 on the wire; when
    gather_dim != 0 a post-op re-lays the result (chunk+cat) — the fused
    HIP relayout kernel replaces that cat on GPU."""
    n = _world(group)
    t = t.contiguous()
    out = torch.empty((n * t.shape[0],) + tuple(t.shape[1:]), dtype=t.dtype,
                      device=t.device)
    work = dist.all_gather_into_tensor(out, t, group=group, async_op=True)
    post = None
    if gather_dim != 0:
        def post(res):
            return torch.cat(torch.chunk(res, n, dim=0), dim=gather_dim)
    return _Work(out, work, post)


def all_gather(t, gather_dim: int = 0, group=None):
    return comm_wait(all_gather_start(t, gather_dim, group))


# -------------------------------------------------------- reduce_scatter -----
def reduce_scatter_start(t: torch.Tensor, scatter_dim: int = 0,
                         op: str = "sum", group=None) -> _Work:
    """PARTIAL -> SHARD(dim)."""
    n = _world(group)
    if scatter_dim != 0:
        t = torch.cat(torch.chunk(t, n, dim=scatter_dim), dim=0)
    t = t.contiguous()
    if _is_gloo(group):
        # gloo has no reduce_scatter_tensor: all_reduce then slice
        red = dist.ReduceOp.SUM
        work = dist.all_reduce(t, op=red, group=group, async_op=True)
        rank = _rank(group)

        def post(res):
            piece = torch.chunk(res, n, dim=0)[rank]
            if op == "avg":
                piece = piece / n
            return piece.contiguous()
        return _Work(t, work, post)
    out = torch.empty((t.shape[0] // n,) + tuple(t.shape[1:]), dtype=t.dtype,
                      device=t.device)
    work = dist.reduce_scatter_tensor(out, t, op=REDUCE_OPS[op], group=group,
                                      async_op=True)
    return _Work(out, work)


def reduce_scatter(t, scatter_dim: int = 0, op: str = "sum", group=None):
    return comm_wait(reduce_scatter_start(t, scatter_dim, op, group))


# ------------------------------------------------------------ all_to_all -----
def all_to_all_start(t: torch.Tensor, src_dim: int, dst_dim: int,
                     group=None) -> _Work:
    """SHARD(src_dim) -> SHARD(dst_dim): one pairwise exchange over xGMI.

    Local input is the src_dim shard; output is the dst_dim shard. We move
    dst_dim chunks to their owners with `all_to_all_single` (RCCL pairwise
    p2p over the 7 xGMI links), then re-lay.
    """
    n = _world(group)
    assert src_dim != dst_dim
    # split local tensor into n chunks along dst_dim, send chunk i to rank i
    send = torch.cat(torch.chunk(t, n, dim=dst_dim), dim=0).contiguous()
    if _is_gloo(group):
        # emulate: all_gather then select my dst chunk
        gout = torch.empty((n * send.shape[0],) + tuple(send.shape[1:]),
                           dtype=send.dtype, device=send.device)
        work = dist.all_gather_into_tensor(gout, send, group=group,
                                           async_op=True)
        rank = _rank(group)

        def post(res):
            # res: [n_src * n_dst * local0] — pick dst block == my rank from
            # every source, cat along src_dim
            blocks = torch.chunk(res, n * n, dim=0)
            mine = [blocks[s * n + rank] for s in range(n)]
            return torch.cat(mine, dim=src_dim).contiguous()
        return _Work(gout, work, post)
    out = torch.empty_like(send)
    work = dist.all_to_all_single(out, send, group=group, async_op=True)

    def post(res):
        return torch.cat(torch.chunk(res, n, dim=0), dim=src_dim).contiguous()
    return _Work(out, work, post)


def all_to_all(t, src_dim: int, dst_dim: int, group=None):
    return comm_wait(all_to_all_start(t, src_dim, dst_dim, group))


# ------------------------------------------------------------- local ops -----
def local_chunk(t: torch.Tensor, dim: int = 0, group=None) -> torch.Tensor:
    """REPLICATE -> SHARD(dim): no communication, slice my piece."""
    n = _world(group)
    rank = _rank(group)
    return torch.chunk(t, n, dim=dim)[rank].contiguous()


def partial_localize(t: torch.Tensor, group=None) -> torch.Tensor:
    """REPLICATE -> PARTIAL(sum): keep value on rank 0, zeros elsewhere."""
    if _rank(group) == 0:
        return t
    return torch.zeros_like(t)


# -------------------------------------------------------- expert parallel ----
def _all_to_all_ep_impl(t: torch.Tensor, group) -> torch.Tensor:
    n = _world(group)
    assert t.shape[0] % n == 0
    t = t.contiguous()
    if n == 1:
        return t
    if _is_gloo(group):
        gout = torch.empty((n * t.shape[0],) + tuple(t.shape[1:]),
                           dtype=t.dtype, device=t.device)
        dist.all_gather_into_tensor(gout, t, group=group)
        rank = _rank(group)
        # block (src s, chunk w) at index s*n + w; take w == my rank
        blocks = torch.chunk(gout, n * n, dim=0)
        return torch.cat([blocks[s * n + rank] for s in range(n)],
                         dim=0).contiguous()
    out = torch.empty_like(t)
    dist.all_to_all_single(out, t, group=group)
    return out


class _AllToAllEP(torch.autograd.Function):
    """Autograd-aware equal-split all-to-all: the backward of an
    all-to-all permutation is the same all-to-all on the grads."""

    @staticmethod
    def forward(ctx, t, group):
        ctx.group = group
        with torch.no_grad():
            return _all_to_all_ep_impl(t.detach(), group)

    @staticmethod
    def backward(ctx, grad):
        with torch.no_grad():
            g = _all_to_all_ep_impl(grad.contiguous(), ctx.group)
        return g, None


def all_to_all_ep(t: torch.Tensor, group=None) -> torch.Tensor:
    """Equal-split all-to-all along dim 0 (expert dispatch/combine).

    Input [W*L, ...]: chunk w goes to rank w; output chunk w is what rank
    w sent here. On RCCL this is ONE all_to_all_single = W-1 simultaneous
    pairwise xGMI exchanges; gloo (CPU tests) emulates via all_gather.
    Differentiable (EP training needs grads flowing back through
    dispatch AND combine).
    """
    if torch.is_grad_enabled() and t.requires_grad:
        return _AllToAllEP.apply(t, group)
    return _all_to_all_ep_impl(t, group)


# ---------------------------------------------------- flat (ZeRO) helpers ----
def _flat_pad(t: torch.Tensor, n: int) -> torch.Tensor:
    """Flatten to 1-D and zero-pad so numel % n == 0."""
    flat = t.reshape(-1)
    pad = (-flat.numel()) % n
    if pad:
        flat = torch.cat([flat, flat.new_zeros(pad)])
    return flat.contiguous()


def flat_shard_local(t: torch.Tensor, group=None) -> torch.Tensor:
    """REPLICATE -> FLAT shard: my 1-D slice of the padded flat tensor."""
    n = _world(group)
    flat = _flat_pad(t, n)
    return torch.chunk(flat, n, dim=0)[_rank(group)].contiguous()


def reduce_scatter_flat(t: torch.Tensor, op: str = "avg",
                        group=None) -> torch.Tensor:
    """grad (full, per-rank partial) -> reduced 1-D local shard."""
    n = _world(group)
    return reduce_scatter(_flat_pad(t, n), 0, op, group)


def all_gather_flat(shard: torch.Tensor, shape, group=None) -> torch.Tensor:
    """FLAT shard -> full tensor of `shape` (drops the pad)."""
    full = all_gather(shard.contiguous(), 0, group)
    numel = 1
    for s in shape:
        numel *= s
    return full[:numel].reshape(tuple(shape))


# ---------------------------------------------------------------- PP p2p -----
def batch_p2p(p2p_ops: List[dist.P2POp]):
    if not p2p_ops:
        return []
    return dist.batch_isend_irecv(p2p_ops)


# ------------------------------------------------------ grouped (bucketed) ---
def all_reduce_bucket_start(tensors, op: str = "sum", group=None):
    """ONE flat all-reduce for many small tensors (reference
    comm_optimize.py:356-390 grouped_comm, re-done as a single fused
    wire buffer: per-call RCCL latency is ~10us, so a ZeRO/ddp tail of
    per-param collectives is latency-bound)."""
    shapes = [tuple(t.shape) for t in tensors]
    numels = [t.numel() for t in tensors]
    flat = torch.cat([t.reshape(-1) for t in tensors])
    w = all_reduce_start(flat, op, group)

    def post(res):
        outs = []
        off = 0
        for sh, n in zip(shapes, numels):
            outs.append(res[off:off + n].reshape(sh))
            off += n
        return outs
    base_post = w.post
    w.post = (lambda x: post(base_post(x))) if base_post else post
    return w


def all_gather_bucket_start(tensors, gather_dims=None, group=None):
    """ONE flat all-gather for many tensors (per-tensor gather dim).
    Wire layout: concat of the local shards; each output tensor is
    reassembled from the W rank segments along its own dim."""
    n = _world(group)
    gather_dims = gather_dims or [0] * len(tensors)
    shapes = [tuple(t.shape) for t in tensors]
    numels = [t.numel() for t in tensors]
    total = sum(numels)
    flat = torch.cat([t.contiguous().reshape(-1) for t in tensors])
    out = torch.empty(n * total, dtype=flat.dtype, device=flat.device)
    work = dist.all_gather_into_tensor(out, flat, group=group, async_op=True)

    def post(res):
        outs = []
        off = 0
        for sh, ne, d in zip(shapes, numels, gather_dims):
            parts = [res[r * total + off: r * total + off + ne].reshape(sh)
                     for r in range(n)]
            outs.append(torch.cat(parts, dim=d))
            off += ne
        return outs
    return _Work(out, work, post)


# -------------------------------------------------------- P2P reshard -------
def _rect_of(shape, placements, coords, mesh_shape):
    """Global index rectangle [(lo, hi), ...] owned by a rank at `coords`
    under a per-mesh-dim placement vector (SHARD/REPLICATE only; outer
    mesh dims chunk first, matching shard_tensor_local)."""
    rect = [[0, s] for s in shape]
    for d, p in enumerate(placements):
        if p[0] != "S":
            continue
        td = p[1]
        lo, hi = rect[td]
        size = (hi - lo) // mesh_shape[d]
        lo = lo + coords[d] * size
        rect[td] = [lo, lo + size]
    return rect


def _intersect(a, b):
    out = []
    for (lo1, hi1), (lo2, hi2) in zip(a, b):
        lo, hi = max(lo1, lo2), min(hi1, hi2)
        if lo >= hi:
            return None
        out.append((lo, hi))
    return out


def p2p_reshard(t: torch.Tensor, global_shape, cur, want, mesh):
    """Rectangle-intersection P2P reshard (reference sharding.py:336-612
    re-designed for xGMI: every pairwise move is one direct link hop).

    cur/want: per-mesh-dim placement tuples ("S", td) | ("R",). Valid when
    both contain only SHARD/REPLICATE; a dim replicated in `cur` picks the
    sender whose coords match the receiver on that mesh dim."""
    md = mesh.mesh
    mesh_shape = tuple(md.shape)
    ranks = md.mesh.flatten().tolist()
    me = dist.get_rank()
    my_idx = ranks.index(me)

    def coords_of(idx):
        c = []
        rem = idx
        for s in reversed(mesh_shape):
            c.append(rem % s)
            rem //= s
        return list(reversed(c))

    my_c = coords_of(my_idx)
    my_cur = _rect_of(global_shape, cur, my_c, mesh_shape)
    my_want = _rect_of(global_shape, want, my_c, mesh_shape)
    out = torch.empty([hi - lo for lo, hi in my_want], dtype=t.dtype,
                      device=t.device)

    def rel(rect, base):
        return tuple(slice(lo - blo, hi - blo)
                     for (lo, hi), (blo, bhi) in zip(rect, base))

    def canonical_pair(sender_c, recv_c):
        # on mesh dims where cur is replicated, the canonical sender has
        # the receiver's coordinate (keeps traffic inside subgroups)
        for d, p in enumerate(cur):
            if p[0] != "S" and sender_c[d] != recv_c[d]:
                return False
        return True

    ops = []
    recv_bufs = []
    t = t.contiguous()
    for idx, r in enumerate(ranks):
        c = coords_of(idx)
        if r != me:
            # what I send to r
            r_want = _rect_of(global_shape, want, c, mesh_shape)
            inter = _intersect(my_cur, r_want)
            if inter is not None and canonical_pair(my_c, c):
                piece = t[rel(inter, my_cur)].contiguous()
                ops.append(dist.P2POp(dist.isend, piece, r))
            # what I receive from r
            r_cur = _rect_of(global_shape, cur, c, mesh_shape)
            inter2 = _intersect(r_cur, my_want)
            if inter2 is not None and canonical_pair(c, my_c):
                buf = torch.empty([hi - lo for lo, hi in inter2],
                                  dtype=t.dtype, device=t.device)
                ops.append(dist.P2POp(dist.irecv, buf, r))
                recv_bufs.append((inter2, buf))
        else:
            inter = _intersect(my_cur, my_want)
            if inter is not None:
                out[rel(inter, my_want)] = t[rel(inter, my_cur)]
    if ops:
        for wk in dist.batch_isend_irecv(ops):
            wk.wait()
    for rect, buf in recv_bufs:
        out[rel(rect, my_want)] = buf
    return out
