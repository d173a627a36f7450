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


def milp_schedule(gm: fx.GraphModule, durations: Dict[str, float],
                  max_tasks: int = 80,
                  time_limit: float = 10.0) -> Optional[List[fx.Node]]:
    """Exact two-resource RCPSP via scipy's HiGHS MILP (capability parity
    with the reference's exact 'general' method, schedule/rcpsp.py:60-180
    there — CP-SAT; here the same disjunctive model as a big-M MILP on
    the solver stack the rest of this codebase already uses).

    Tasks = compute nodes with positive duration (comp resource) and
    comm ``*_start`` nodes (comm resource, duration = transfer time).
    Start times are continuous; same-resource non-ordered pairs get a
    sequencing binary. Minimizes makespan; returns the node list
    stable-sorted by optimal start time (ties by original topo index),
    or None when the graph has no comm window or is too large (the
    caller falls back to odd_even_schedule).
    """
    import numpy as np
    from scipy import sparse
    from scipy.optimize import Bounds, LinearConstraint, milp

    nodes = list(gm.graph.nodes)
    idx_of = {n: i for i, n in enumerate(nodes)}
    starts = [n for n in nodes if n.op == "call_function"
              and n.target in COMM_START_TARGETS]
    if not starts:
        return None

    comp = [n for n in nodes if n.op == "call_function"
            and n.target not in COMM_START_TARGETS
            and n.target is not rt_wait
            and durations.get(n.name, 0.0) > 0.0]
    tasks = comp + starts
    if len(tasks) > max_tasks:
        return None
    t_idx = {n: i for i, n in enumerate(tasks)}
    dur = [max(durations.get(n.name, 0.0), 1e-6) for n in tasks]
    n_t = len(tasks)

    # dependency closure BETWEEN tasks, through any non-task nodes
    preds_of: Dict[fx.Node, set] = {}

    def task_preds(n: fx.Node) -> set:
        if n in preds_of:
            return preds_of[n]
        acc = set()
        for p in n.all_input_nodes:
            if p in t_idx:
                acc.add(p)
            acc |= task_preds(p)
        preds_of[n] = acc
        return acc

    for n in nodes:
        task_preds(n)
    anc = [set() for _ in range(n_t)]           # transitive task ancestors
    for n in tasks:
        work = set(task_preds(n))
        seen = set()
        while work:
            p = work.pop()
            if p in seen:
                continue
            seen.add(p)
            work |= task_preds(p)
        anc[t_idx[n]] = {t_idx[p] for p in seen}

    # vars: [s_0..s_{n-1}, C, y_...] with one y per unordered same-res pair
    pairs = []
    res_of = [0] * len(comp) + [1] * len(starts)
    for i in range(n_t):
        for j in range(i + 1, n_t):
            if res_of[i] != res_of[j]:
                continue
            if i in anc[j] or j in anc[i]:
                continue
            pairs.append((i, j))
    n_var = n_t + 1 + len(pairs)
    H = sum(dur) + 1.0                           # horizon / big-M
    rows, cols, vals, lo, hi = [], [], [], [], []
    r = 0

    def add(coefs, lb, ub):
        nonlocal r
        for c, v in coefs:
            rows.append(r)
            cols.append(c)
            vals.append(v)
        lo.append(lb)
        hi.append(ub)
        r += 1

    for n in tasks:                              # precedence
        j = t_idx[n]
        for p in task_preds(n):
            i = t_idx[p]
            if i == j:
                continue
            add([(j, 1.0), (i, -1.0)], dur[i], np.inf)
    for i in range(n_t):                         # makespan
        add([(n_t, 1.0), (i, -1.0)], dur[i], np.inf)
    for k, (i, j) in enumerate(pairs):           # disjunctive sequencing
        y = n_t + 1 + k
        # y=1 -> i before j:  s_j - s_i - H*y >= d_i - H
        # y=0 -> j before i:  s_i - s_j + H*y >= d_j
        add([(j, 1.0), (i, -1.0), (y, -H)], dur[i] - H, np.inf)
        add([(i, 1.0), (j, -1.0), (y, H)], dur[j], np.inf)

    A = sparse.csr_matrix((vals, (rows, cols)), shape=(r, n_var))
    c = np.zeros(n_var)
    c[n_t] = 1.0
    integrality = np.zeros(n_var)
    integrality[n_t + 1:] = 1
    bounds = Bounds(lb=np.zeros(n_var),
                    ub=np.concatenate([np.full(n_t + 1, H),
                                       np.ones(len(pairs))]))
    res = milp(c=c, constraints=LinearConstraint(A, lo, hi),
               integrality=integrality, bounds=bounds,
               options={"time_limit": time_limit})
    if res.x is None:
        return None

    s_task = {tasks[i]: float(res.x[i]) for i in range(n_t)}
    # propagate completion times to every node, stable-sort by (time, idx)
    t_of: Dict[fx.Node, float] = {}
    for n in nodes:
        if n.op == "output":
            t_of[n] = float("inf")   # fx requires the output node last
        elif n in s_task:
            t_of[n] = s_task[n]
        else:
            t_of[n] = max((t_of[p] + durations.get(p.name, 0.0)
                           for p in n.all_input_nodes), default=0.0)
    order = sorted(nodes, key=lambda n: (t_of[n], idx_of[n]))
    return order
