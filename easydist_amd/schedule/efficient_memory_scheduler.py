"""Min-skyline static address packing for the HIP runtime allocator.

Capability parity with reference ``easydist/torch/schedule/
efficient_memory_scheduler.py`` (gen_mem_addresses, 32-405): pack every
profiled buffer at a fixed arena offset such that buffers with
overlapping lifetimes never overlap in address space; emit the plan in
MALLOC ORDER (the order the graph's allocations arrive at the allocator),
which the C++ RUNTIME mode plays back.

An ILP variant (reference ilp_memory_scheduler.py / MODeL) can be layered
on the same Buffer list; the skyline heuristic is the default there too.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Tuple

import torch.fx as fx

from ..memory.allocator_profiler import GraphMemInfo
from .lifetime import Buffer, build_lifetimes

logger = logging.getLogger(__name__)

ALIGN = 512   # byte alignment for every planned buffer


def _align(x: int) -> int:
    return (x + ALIGN - 1) // ALIGN * ALIGN


def pack_buffers(buffers: List[Buffer]) -> Tuple[Dict[Tuple[str, int], int],
                                                 int]:
    """Greedy min-offset packing: process buffers in allocation order,
    place each at the lowest aligned offset not overlapping any
    live-interval-overlapping buffer already placed."""
    placed: List[Tuple[int, int, Buffer]] = []   # (offset, size, buf)
    addresses: Dict[Tuple[str, int], int] = {}
    peak = 0
    for b in sorted(buffers, key=lambda b: (b.start, -b.size)):
        overlapping = [(off, sz) for off, sz, other in placed
                       if not (other.end < b.start or b.end < other.start)]
        overlapping.sort()
        candidate = 0
        for off, sz in overlapping:
            if candidate + b.size <= off:
                break
            candidate = max(candidate, _align(off + sz))
        addresses[(b.node_name, b.alloc_idx)] = candidate
        placed.append((candidate, b.size, b))
        peak = max(peak, candidate + b.size)
    return addresses, peak


def plan_from_events(events):
    """Build the plan from the RECORDED alloc/free event sequence of one
    real execution of the step (the authoritative lifetime source: torch's
    pluggable-allocator metadata requires that a planned address is never
    re-served before the tensor that held it was actually freed).

    events: [(is_alloc, ptr, size)] in wall order. Returns
    (entries, arena_size, stats) — entries in MALLOC order."""
    live: Dict[int, int] = {}          # ptr -> alloc index
    intervals: List[Buffer] = []
    streams: List[int] = []
    alloc_count = 0
    n_events = len(events)
    # stream-aware: lifetimes below assume the event
    # ORDER reflects execution order, which only holds within one
    # stream. The dominant (compute) stream is planned into the arena;
    # side-stream allocations keep their own plan sequence but are
    # served by the backing allocator (offset -1) — their cross-stream
    # interleave is not deterministic.
    from collections import Counter
    scount = Counter(e[3] if len(e) > 3 else 0
                     for e in events if e[0] == 1)
    main_stream = scount.most_common(1)[0][0] if scount else 0
    for t, ev in enumerate(events):
        is_alloc, ptr, size = ev[0], ev[1], ev[2]
        stream = ev[3] if len(ev) > 3 else 0
        if is_alloc:
            live[ptr] = alloc_count
            intervals.append(Buffer(f"a{alloc_count}", 0, size, t,
                                    n_events, False))
            streams.append(stream)
            alloc_count += 1
        else:
            i = live.pop(ptr, None)
            if i is not None:
                intervals[i].end = t - 1   # freed AT t: reusable from t
    main = [b for b, st in zip(intervals, streams) if st == main_stream]
    addresses, peak = pack_buffers(main)
    entries = [(addresses[(b.node_name, 0)] if st == main_stream else -1,
                b.size, st)
               for b, st in zip(intervals, streams)]
    naive = sum(b.size for b in intervals)
    stats = {"arena_bytes": peak, "naive_sum_bytes": naive,
             "n_allocs": len(entries),
             "savings": 1.0 - (peak / naive if naive else 0.0)}
    logger.info("memory plan (events): %d allocs, arena %.1f MiB vs naive "
                "%.1f MiB (%.0f%% saved)", len(entries), peak / 2**20,
                naive / 2**20, stats["savings"] * 100)
    return entries, peak, stats


def plan_memory(gm: fx.GraphModule, mem_info: GraphMemInfo):
    """Returns (plan_entries, arena_size, stats).

    plan_entries: [(offset, size)] in malloc order — offset -1 delegates
    that allocation to the backing allocator (never planned).
    """
    buffers = build_lifetimes(gm, mem_info)
    addresses, peak = pack_buffers(buffers)
    entries: List[Tuple[int, int]] = []
    naive = 0
    for node_name, alloc_idx, size in mem_info.alloc_order:
        off = addresses.get((node_name, alloc_idx), -1)
        entries.append((off, size))
        naive += size
    stats = {"arena_bytes": peak, "naive_sum_bytes": naive,
             "n_allocs": len(entries),
             "savings": 1.0 - (peak / naive if naive else 0.0)}
    logger.info("memory plan: %d allocs, arena %.1f MiB vs naive %.1f MiB "
                "(%.0f%% saved)", len(entries), peak / 2**20, naive / 2**20,
                stats["savings"] * 100)
    return entries, peak, stats


def plot_plan(buffers, addresses, peak, path):
    """Address-vs-time rectangle plot of a packed plan (reference
    observability: efficient_memory_scheduler.py:385-405 matplotlib
    memory plots). Each buffer is a rectangle [start,end] x
    [offset, offset+size]."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    from matplotlib.patches import Rectangle

    fig, ax = plt.subplots(figsize=(10, 6))
    for b in buffers:
        off = addresses.get((b.node_name, b.alloc_idx))
        if off is None:
            continue
        ax.add_patch(Rectangle((b.start, off / 2**20),
                               max(b.end - b.start, 0.5),
                               b.size / 2**20,
                               alpha=0.6, edgecolor="black",
                               linewidth=0.3))
    ax.set_xlim(0, max((b.end for b in buffers), default=1) + 1)
    ax.set_ylim(0, peak / 2**20 * 1.05)
    ax.set_xlabel("op index (lifetime)")
    ax.set_ylabel("arena offset (MiB)")
    ax.set_title(f"static memory plan — peak {peak / 2**20:.1f} MiB, "
                 f"{len(buffers)} buffers")
    fig.tight_layout()
    fig.savefig(path, dpi=110)
    plt.close(fig)
