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
    if durations:
        from ...schedule.rcpsp import odd_even_schedule
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
