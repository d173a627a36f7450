"""Communication overlap passes on the sharded graph.

Capability parity with reference ``easydist/torch/passes/comm_optimize.py``
(comm_optimize + grouped_comm, lines 288-410) re-designed for the
MI355X runtime: RCCL launches collectives on their own HIP streams, so
overlap is created purely by SCHEDULING — issue every ``*_start`` as
early as its input allows and sink every ``rt_wait`` to just before its
first consumer. The window between them runs compute concurrently with
the xGMI transfer. Op ordering beyond that is delegated to the RCPSP
odd-even heuristic (schedule/rcpsp.py, reference schedule/rcpsp.py:
276-330) when durations are available from the runtime profiler.
"""
from __future__ import annotations

import logging
import operator
from typing import Dict, List, Optional

import torch.fx as fx

from ...runtime.comm_runtime import COMM_START_TARGETS, rt_wait

logger = logging.getLogger(__name__)


def _is_start(n: fx.Node) -> bool:
    return n.op == "call_function" and n.target in COMM_START_TARGETS


def _is_wait(n: fx.Node) -> bool:
    return n.op == "call_function" and n.target is rt_wait


def sink_waits_raise_starts(gm: fx.GraphModule) -> int:
    """Maximize each collective's overlap window. Returns #moved nodes."""
    graph = gm.graph
    moved = 0

    # One order snapshot per phase (O(n)), not one per move (O(n^2) on the
    # multi-thousand-node whole-step traces this runs on). Moving a start
    # never reorders two NON-moved nodes relative to each other, so stale
    # indices are only a correctness risk when a start's producer is
    # itself a start moved earlier in this phase — rebuild just then.
    # Adjacency ("already in place") is read off the node linked list.
    moved_this_phase: set = set()

    # raise starts: insert each start right after its last-placed producer
    order = {n: i for i, n in enumerate(graph.nodes)}
    for n in list(graph.nodes):
        if not _is_start(n):
            continue
        producers = [a for a in n.all_input_nodes]
        if not producers:
            continue
        if moved_this_phase.intersection(producers):
            order = {x: i for i, x in enumerate(graph.nodes)}
            moved_this_phase.clear()
        anchor = max(producers, key=lambda p: order[p])
        if anchor.next is n:      # already directly after its producer
            continue
        anchor.append(n)          # move n to directly after anchor
        moved += 1
        moved_this_phase.add(n)

    # sink waits: place each wait right before its first consumer. Waits
    # are never another wait's input, so one snapshot suffices.
    order = {n: i for i, n in enumerate(graph.nodes)}
    for n in list(graph.nodes):
        if not _is_wait(n):
            continue
        users = [u for u in n.users]
        if not users:
            continue
        first = min(users, key=lambda u: order[u])
        if n.next is first:       # already directly before first consumer
            continue
        first.prepend(n)
        moved += 1

    if moved:
        graph.lint()
        gm.recompile()
        logger.info("comm_optimize: repositioned %d comm nodes", moved)
    return moved


def comm_cse(gm: fx.GraphModule) -> int:
    """Eliminate duplicate collectives on identical SSA values.

    The sharded graph is functional: two ``rt_*_start`` nodes with the
    SAME input node and args move the same bytes twice (typical case: a
    ZeRO-style weight all-gather issued in forward AND again in
    backward). Keep the first start/wait pair, point later duplicates'
    waits at the first wait's result. Lifetime grows (the gathered
    tensor stays live between uses) but every duplicated transfer and
    launch disappears — the right trade on xGMI where per-link ring
    bandwidth, not memory, bounds the step."""
    graph = gm.graph
    seen = {}
    removed = 0
    for n in list(graph.nodes):
        if not _is_start(n):
            continue
        key = (str(n.target), tuple(
            a.name if isinstance(a, fx.Node) else repr(a) for a in n.args))
        waits = [u for u in n.users if _is_wait(u)]
        if len(waits) != 1 or len(n.users) != 1:
            continue
        if key in seen:
            first_wait = seen[key]
            waits[0].replace_all_uses_with(first_wait)
            graph.erase_node(waits[0])
            graph.erase_node(n)
            removed += 1
        else:
            seen[key] = waits[0]
    if removed:
        graph.lint()
        gm.recompile()
        logger.info("comm_cse: removed %d duplicate collectives", removed)
    return removed


def comm_optimize(gm: fx.GraphModule, durations: Optional[Dict[str, float]]
                  = None, method: str = "odd_even") -> fx.GraphModule:
    """Entry point mirroring the reference's comm_optimize: reposition
    start/wait pairs; when per-node durations are available, additionally
    reorder independent compute between start/wait pairs with the RCPSP
    odd-even heuristic."""
    sink_waits_raise_starts(gm)
    group_collectives(gm)
    if durations:
        from ...schedule.rcpsp import milp_schedule, odd_even_schedule
        order = None
        if method == "milp":
            # exact disjunctive MILP for small graphs; None on fallthrough
            order = milp_schedule(gm, durations)
        if order is None:
            order = odd_even_schedule(gm, durations)
        if order is not None:
            _relink(gm, order)
    return gm


def _relink(gm: fx.GraphModule, order: List[fx.Node]):
    """Rebuild the node list in the given (topologically valid) order."""
    graph = gm.graph
    anchor = None
    for n in order:
        if anchor is None:
            anchor = n
            continue
        anchor.append(n)
        anchor = n
    graph.lint()
    gm.recompile()


def group_collectives(gm: fx.GraphModule, min_group: int = 2,
                      max_bucket_bytes: int = 64 << 20) -> int:
    """Fuse compatible small collectives into ONE flat bucketed call
    (reference comm_optimize.py:356-390 grouped_comm).

    Run construction: walk candidate (start, wait) pairs in graph order,
    accumulating while (a) no already-collected wait has a consumer that
    precedes the next start (fusing would need that result before the
    bucket executes — a cycle), and (b) the bucket stays under the size
    cap. The grouped start+wait land at the LAST member start's position
    (every member's input precedes its own start, hence the bucket).
    all_gather members carry per-tensor gather dims (ZeRO re-gathers
    updated param shards along the dim the solver picked).
    """
    from ...runtime.comm_runtime import (rt_all_gather_start,
                                         rt_all_reduce_start,
                                         rt_grouped_all_gather_start,
                                         rt_grouped_all_reduce_start,
                                         rt_grouped_wait)
    graph = gm.graph
    order = {n: i for i, n in enumerate(graph.nodes)}

    def nbytes(n):
        v = n.meta.get("val") if hasattr(n, "meta") else None
        import torch as _t
        return (v.numel() * v.element_size()
                if isinstance(v, _t.Tensor) else 1 << 30)

    groups = {}
    for n in graph.nodes:
        if n.op != "call_function":
            continue
        if n.target is rt_all_reduce_start:
            key = ("ar", n.args[1], n.args[2])
        elif n.target is rt_all_gather_start:
            key = ("ag", n.args[2])
        else:
            continue
        waits = [u for u in n.users if _is_wait(u)]
        if len(waits) != 1 or len(n.users) != 1:
            continue
        groups.setdefault(key, []).append((n, waits[0]))

    n_grouped = 0
    INF = float("inf")
    for key, pairs in groups.items():
        if len(pairs) < min_group:
            continue
        pairs.sort(key=lambda p: order[p[0]])
        runs = []
        cur, cur_bytes, barrier = [], 0, INF
        for s, w in pairs:
            b = nbytes(s.args[0])
            if cur and (order[s] > barrier
                        or cur_bytes + b > max_bucket_bytes):
                runs.append(cur)
                cur, cur_bytes, barrier = [], 0, INF
            cur.append((s, w))
            cur_bytes += b
            u_min = min((order[u] for u in w.users), default=INF)
            barrier = min(barrier, u_min)
        if cur:
            runs.append(cur)
        for run in runs:
            if len(run) < min_group:
                continue
            starts = [s for s, _ in run]
            waits = [w for _, w in run]
            last_s = starts[-1]
            with graph.inserting_before(last_s):
                if key[0] == "ar":
                    gs = graph.call_function(
                        rt_grouped_all_reduce_start,
                        ([s.args[0] for s in starts], key[1], key[2]))
                else:
                    gs = graph.call_function(
                        rt_grouped_all_gather_start,
                        ([s.args[0] for s in starts],
                         [s.args[1] for s in starts], key[1]))
                gw = graph.call_function(rt_grouped_wait, (gs,))
                items = [graph.call_function(operator.getitem, (gw, i))
                         for i in range(len(run))]
            for (s, w), item in zip(run, items):
                item.meta = dict(w.meta)
                w.replace_all_uses_with(item)
                graph.erase_node(w)
                graph.erase_node(s)
            n_grouped += len(run)

    if n_grouped:
        graph.lint()
        gm.recompile()
        logger.info("group_collectives: fused %d collectives into buckets",
                    n_grouped)
    return n_grouped
