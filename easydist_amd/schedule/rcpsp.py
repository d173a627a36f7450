"""RCPSP op scheduling: odd-even block-swap heuristic.

Capability parity with reference ``easydist/torch/schedule/rcpsp.py``
(default 'odd_even' method, lines 276-330; the CP-SAT/'general' and
MIP/'genetic' methods there need ortools, which this image does not
carry — the heuristic is the production default in the reference too).

Model: two resources {comp, comm}. A comm interval is a (start, wait)
pair whose transfer occupies the comm resource; compute ops occupy comp.
The heuristic walks adjacent (compute, compute) pairs between a
start/wait window and swaps them when moving a LONGER independent
compute op into the window increases overlap with the transfer.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch.fx as fx

from ..runtime.comm_runtime import COMM_START_TARGETS, rt_wait

logger = logging.getLogger(__name__)


def _deps_ok(order: List[fx.Node], i: int, j: int) -> bool:
    """Can order[i] and order[j] swap (i<j adjacent)? Only if j doesn't
    consume i."""
    a, b = order[i], order[j]
    return a not in b.all_input_nodes


def odd_even_schedule(gm: fx.GraphModule, durations: Dict[str, float],
                      max_rounds: int = 8) -> Optional[List[fx.Node]]:
    """Return a reordered node list (or None if no comm windows exist)."""
    nodes = list(gm.graph.nodes)
    starts = [n for n in nodes if n.op == "call_function"
              and n.target in COMM_START_TARGETS]
    if not starts:
        return None
    waits = {n for n in nodes if n.op == "call_function"
             and n.target is rt_wait}

    order = nodes[:]
    pos = {n: i for i, n in enumerate(order)}

    def window_of(start: fx.Node):
        """Indices (s, w) of the start and its wait."""
        w = next((u for u in start.users if u in waits), None)
        return (pos[start], pos[w]) if w is not None else (pos[start], None)

    changed = True
    rounds = 0
    while changed and rounds < max_rounds:
        changed = False
        rounds += 1
        pos = {n: i for i, n in enumerate(order)}
        for start in starts:
            s, w = window_of(start)
            if w is None or w - s > 8:
                continue   # window already wide
            # try to pull the next independent compute op (after the wait)
            # into the window, pushing the wait later
            j = w + 1
            while j < len(order):
                cand = order[j]
                if cand.op != "call_function" or cand in waits or \
                        cand.target in COMM_START_TARGETS:
                    break
                wait_node = order[w]
                if wait_node in cand.all_input_nodes:
                    break   # depends on the transfer: cannot cross
                dur = durations.get(cand.name, 0.0)
                if dur <= 0:
                    break
                # move cand before the wait
                order.pop(j)
                order.insert(w, cand)
                pos = {n: i for i, n in enumerate(order)}
                w += 1
                j = w + 1
                changed = True
    return order
