"""tile_comm: split big collectives into chunks so transfer and compute
pipeline within one op's window.

Capability parity with reference ``easydist/torch/passes/tile_comm.py``
(36-100, 403+) re-designed for the RCCL runtime: a collective larger than
``tile_threshold_bytes`` is rewritten into T chunked start/wait pairs
along dim 0 plus a cat — RCCL runs each chunk on its collective stream,
so chunk i's transfer overlaps chunk i-1's consumer compute once the
sink/raise pass spreads the waits. (The reference additionally tiles the
producer/consumer matmuls; on MI355X the hipBLASLt GEMMs are already
stream-pipelined against RCCL, so graph-level comm tiling captures the
win without rewriting compute.)
"""
from __future__ import annotations

import logging
import operator
from typing import List

import torch
import torch.fx as fx

from ... import config as mdconfig
from ...runtime import comm_runtime as crt

logger = logging.getLogger(__name__)

# all_reduce only: it is elementwise so row-tiling + cat reproduces the
# result; tiled all_gather/reduce_scatter would permute the rank-major
# row order and need a re-interleave (not worth the extra copy)
TILABLE = {crt.rt_all_reduce_start}


def _nbytes(n: fx.Node) -> int:
    v = n.meta.get("val") if hasattr(n, "meta") else None
    if isinstance(v, torch.Tensor):
        return v.numel() * v.element_size()
    return 0


def tile_comm(gm: fx.GraphModule, n_tiles: int = 4,
              threshold_bytes: int = 16 << 20) -> int:
    """Split qualifying collectives; returns #tiled. Chunking is along
    dim 0 and requires divisibility."""
    graph = gm.graph
    tiled = 0
    for n in list(graph.nodes):
        if n.op != "call_function" or n.target not in TILABLE:
            continue
        arg = n.args[0]
        v = arg.meta.get("val") if hasattr(arg, "meta") else None
        if not isinstance(v, torch.Tensor) or v.ndim < 1 \
                or v.shape[0] % n_tiles or _nbytes(arg) < threshold_bytes:
            continue
        waits = [u for u in n.users
                 if u.op == "call_function" and u.target is crt.rt_wait]
        if len(waits) != 1:
            continue
        wait = waits[0]
        rest = n.args[1:]
        with graph.inserting_before(n):
            chunks = graph.call_function(torch.chunk, (arg, n_tiles, 0))
            parts = []
            for t in range(n_tiles):
                c = graph.call_function(operator.getitem, (chunks, t))
                s = graph.call_function(n.target, (c, *rest))
                w = graph.call_function(crt.rt_wait, (s,))
                parts.append(w)
            out = graph.call_function(torch.cat, (parts, 0))
        wait.replace_all_uses_with(out)
        graph.erase_node(wait)
        graph.erase_node(n)
        tiled += 1
    if tiled:
        graph.lint()
        gm.recompile()
        logger.info("tile_comm: tiled %d collectives into %d chunks each",
                    tiled, n_tiles)
    return tiled
