"""AutoFlow strategy solver: per-mesh-dim MILP over cluster strategies.

Capability parity with reference ``easydist/autoflow/solver.py``
(AutoFlowSolver1D: binary strategy vars, linearized edge products, comm+mem
objective, beam-search fallback; reference lines 224-890). Re-designed:

* the MILP runs on scipy's HiGHS (`scipy.optimize.milp`) — no external CBC;
* edge costs come from the xGMI cost model (cost_model.py), not generic
  alpha-free terms;
* linearization needs only a one-sided bound (y >= x_u + x_v - 1, y >= 0)
  because all edge costs are non-negative and the objective minimizes.
"""
from __future__ import annotations

import logging
import time
from typing import Dict, List, Optional, Tuple

import numpy as np

from .. import config as mdconfig
from ..metashard.metair import (ClusterStrategy, MetaGraph, MetaNodeCluster,
                                R, SPMD)
from .cost_model import reshard_cost

logger = logging.getLogger(__name__)

MEM_EPS = 1e-13   # seconds per byte: tie-break toward sharded memory
                  # (small vs comm: an all-gather of B bytes ~ B * 6.5e-12 s)


class AutoFlowSolver1D:
    """Pick one strategy per cluster for ONE mesh dimension."""

    def __init__(self, graph: MetaGraph, mesh_size: int,
                 already_sharded: Optional[Dict[str, Dict[int, int]]] = None,
                 output_constraints: Optional[Dict[str, object]] = None):
        self.graph = graph
        self.mesh_size = mesh_size
        self.already_sharded = already_sharded or {}
        # var name -> 'replicate' | ('follow', input_var_name)
        self.output_constraints = output_constraints or {}
        self.clusters: List[MetaNodeCluster] = []
        self.var_producer_cluster: Dict[str, MetaNodeCluster] = {}

    def add_coarsen_graph(self, clusters: List[MetaNodeCluster]):
        self.clusters = clusters
        for c in clusters:
            c.finalize(self.mesh_size, self.already_sharded)
            for st in c.strategies:
                for v in st.out_placements:
                    self.var_producer_cluster[v] = c
        return self

    # ------------------------------------------------------------ edges ------
    def _edges(self):
        """Yield (producer_cluster, consumer_cluster, var_name, nbytes)."""
        seen = set()
        for cc in self.clusters:
            in_vars = set()
            for st in cc.strategies:
                in_vars.update(st.in_placements.keys())
            # sorted: set iteration is PYTHONHASHSEED-ordered — edge order
            # must be identical across processes/runs or MILP tie-breaking
            # diverges
            for v in sorted(in_vars):
                cp = self.var_producer_cluster.get(v)
                if cp is None or cp is cc:
                    continue
                key = (cp.name, cc.name, v)
                if key in seen:
                    continue
                seen.add(key)
                nbytes = self.graph.vars[v].nbytes if v in self.graph.vars else 0
                yield cp, cc, v, nbytes

    def _edge_cost_matrix(self, cp, cc, v, nbytes):
        m = np.zeros((len(cp.strategies), len(cc.strategies)))
        for i, su in enumerate(cp.strategies):
            pu = su.out_placements.get(v, R)
            for j, sv in enumerate(cc.strategies):
                pv = sv.in_placements.get(v)
                if pv is None:
                    continue
                m[i, j] = reshard_cost(pu, pv, nbytes, self.mesh_size)
        return m

    def _unary_costs(self, c: MetaNodeCluster):
        """Terminal costs: outputs forced REPLICATE / tied to input placement."""
        costs = np.zeros(len(c.strategies))
        for i, st in enumerate(c.strategies):
            costs[i] += st.mem_cost * MEM_EPS + st.comp_cost
            for v, pl in st.out_placements.items():
                con = self.output_constraints.get(v)
                if con is None:
                    continue
                nbytes = self.graph.vars[v].nbytes if v in self.graph.vars else 0
                if con == "replicate":
                    costs[i] += reshard_cost(pl, R, nbytes, self.mesh_size)
                elif isinstance(con, tuple) and con[0] == "follow":
                    # soft: handled as an edge to the placeholder cluster below
                    pass
        return costs

    def _follow_edges(self):
        """Edges tying a state OUTPUT var's placement to its input placeholder."""
        for c in self.clusters:
            for v_out in {v for st in c.strategies for v in st.out_placements}:
                con = self.output_constraints.get(v_out)
                if not (isinstance(con, tuple) and con[0] == "follow"):
                    continue
                v_in = con[1]
                cp = self.var_producer_cluster.get(v_in)
                if cp is None or cp is c:
                    continue
                nbytes = self.graph.vars[v_out].nbytes if v_out in self.graph.vars else 0
                m = np.zeros((len(cp.strategies), len(c.strategies)))
                for i, su in enumerate(cp.strategies):
                    pu = su.out_placements.get(v_in, R)
                    for j, sv in enumerate(c.strategies):
                        pv = sv.out_placements.get(v_out, R)
                        m[i, j] = reshard_cost(pv, pu, nbytes, self.mesh_size)
                yield cp, c, m

    # ------------------------------------------------------------- solve -----
    def ilp_solve(self) -> Dict[str, ClusterStrategy]:
        try:
            return self._ilp_solve_impl()
        except Exception as e:
            logger.warning("MILP failed (%s); falling back to beam search", e)
            return self.beam_search()

    def _ilp_solve_impl(self) -> Dict[str, ClusterStrategy]:
        from scipy import sparse
        from scipy.optimize import Bounds, LinearConstraint, milp

        t0 = time.time()
        # variable layout: x vars per (cluster, strategy), then y per costly pair
        x_index: Dict[Tuple[str, int], int] = {}
        obj: List[float] = []
        integrality: List[int] = []
        for c in self.clusters:
            unary = self._unary_costs(c)
            for s in range(len(c.strategies)):
                x_index[(c.name, s)] = len(obj)
                obj.append(float(unary[s]))
                integrality.append(1)

        rows, cols, vals, rhs_ub = [], [], [], []   # A_ub x <= b_ub
        n_ub = 0

        def add_pair_var(cu_name, i, cv_name, j, cost):
            nonlocal n_ub
            yi = len(obj)
            obj.append(float(cost))
            integrality.append(0)
            # x_u + x_v - y <= 1
            rows.extend([n_ub, n_ub, n_ub])
            cols.extend([x_index[(cu_name, i)], x_index[(cv_name, j)], yi])
            vals.extend([1.0, 1.0, -1.0])
            rhs_ub.append(1.0)
            n_ub += 1

        edges = list(self._edges())
        for cp, cc, v, nbytes in edges:
            m = self._edge_cost_matrix(cp, cc, v, nbytes)
            for i in range(m.shape[0]):
                for j in range(m.shape[1]):
                    if m[i, j] > 0:
                        add_pair_var(cp.name, i, cc.name, j, m[i, j])
        for cp, cc, m in self._follow_edges():
            for i in range(m.shape[0]):
                for j in range(m.shape[1]):
                    if m[i, j] > 0:
                        add_pair_var(cp.name, i, cc.name, j, m[i, j])

        nvar = len(obj)
        constraints = []
        if n_ub:
            A_ub = sparse.csr_matrix((vals, (rows, cols)), shape=(n_ub, nvar))
            constraints.append(LinearConstraint(A_ub, -np.inf, rhs_ub))
        # sum_s x[c,s] == 1
        r2, c2, v2 = [], [], []
        for ci, c in enumerate(self.clusters):
            for s in range(len(c.strategies)):
                r2.append(ci)
                c2.append(x_index[(c.name, s)])
                v2.append(1.0)
        A_eq = sparse.csr_matrix((v2, (r2, c2)), shape=(len(self.clusters), nvar))
        constraints.append(LinearConstraint(A_eq, 1.0, 1.0))

        res = milp(c=np.array(obj), constraints=constraints,
                   integrality=np.array(integrality),
                   bounds=Bounds(0, 1),
                   options={"time_limit": mdconfig.get_solver_time_limit(),
                            "mip_rel_gap": 1e-4})
        if res.x is None:
            raise RuntimeError(f"milp: {res.message}")
        assign: Dict[str, int] = {}
        for c in self.clusters:
            best_s, best_v = 0, -1.0
            for s in range(len(c.strategies)):
                xv = res.x[x_index[(c.name, s)]]
                if xv > best_v:
                    best_v, best_s = xv, s
            assign[c.name] = best_s
        # local refinement: a no-op on converged solves, but repairs a
        # time-limited incumbent (monotone, deterministic)
        assign, refined = self._refine(assign)
        logger.info("AutoFlow MILP: %d clusters, %d vars, %.2fs, obj=%.3e "
                    "(refined %.3e)", len(self.clusters), nvar,
                    time.time() - t0, res.fun, refined)
        return {c.name: c.strategies[assign[c.name]]
                for c in self.clusters}

    # --------------------------------------------------------- beam search ---
    def beam_search(self) -> Dict[str, ClusterStrategy]:
        """Beam over clusters in topo order.

        Candidates are SCORED against the parent's var-placement dict
        (no allocation); dicts are materialized only for the `width`
        survivors of each step — the naive per-candidate dict copies
        made this O(clusters² · width) and slower than the timed-out
        MILP it is supposed to replace."""
        width = mdconfig.beam_width
        # beam entries: (cost, assign dict, var placements dict)
        beam: List[Tuple[float, Dict[str, int], Dict[str, SPMD]]] = \
            [(0.0, {}, {})]
        var_bytes = {name: v.nbytes for name, v in self.graph.vars.items()}
        for c in self.clusters:
            unary = self._unary_costs(c)
            cands = []            # (new_cost, parent_idx, strategy_idx)
            for pi, (cost, _assign, var_pl) in enumerate(beam):
                get = var_pl.get
                for s, st in enumerate(c.strategies):
                    add = float(unary[s])
                    for v, need in st.in_placements.items():
                        have = get(v)
                        if have is not None:
                            add += reshard_cost(have, need,
                                                var_bytes.get(v, 0),
                                                self.mesh_size)
                    cands.append((cost + add, pi, s))
            cands.sort(key=lambda t: t[0])
            new_beam = []
            for cost, pi, s in cands[:width]:
                _pc, passign, pvar = beam[pi]
                na = dict(passign)
                na[c.name] = s
                npl = dict(pvar)
                npl.update(c.strategies[s].out_placements)
                new_beam.append((cost, na, npl))
            beam = new_beam
        best = beam[0]
        assign = {c.name: best[1][c.name] for c in self.clusters}
        assign, final_cost = self._refine(assign)
        logger.info("beam search: %d clusters, width %d, cost %.3e "
                    "(refined %.3e)", len(self.clusters), width, best[0],
                    final_cost)
        return {c.name: c.strategies[assign[c.name]]
                for c in self.clusters}

    def _refine(self, assign):
        """Deterministic coordinate descent over the beam assignment:
        re-pick each cluster's argmin strategy holding its neighbors
        fixed (sweeps in topo order until a fixpoint). Repairs the beam's
        greedy myopia; monotonically non-increasing objective, identical
        on every rank."""
        from collections import defaultdict
        cl_by_name = {c.name: c for c in self.clusters}
        touching = defaultdict(list)   # cluster name -> (cp, cc, v, nbytes)
        for e in self._edges():
            touching[e[0].name].append(e)
            touching[e[1].name].append(e)
        follow = defaultdict(list)     # cluster name -> (cp, cc, m, side)
        for cp, cc, m in self._follow_edges():
            follow[cp.name].append((cp, cc, m, 0))
            follow[cc.name].append((cp, cc, m, 1))
        unary = {c.name: self._unary_costs(c) for c in self.clusters}
        var_bytes = {name: v.nbytes for name, v in self.graph.vars.items()}

        def local_cost(c, s):
            cost = float(unary[c.name][s])
            for cp, cc, v, nb in touching[c.name]:
                if cp.name == c.name:
                    su, sv = c.strategies[s], \
                        cc.strategies[assign[cc.name]]
                else:
                    su, sv = cp.strategies[assign[cp.name]], \
                        c.strategies[s]
                pv = sv.in_placements.get(v)
                if pv is not None:
                    cost += reshard_cost(su.out_placements.get(v, R), pv,
                                         nb, self.mesh_size)
            for cp, cc, m, side in follow[c.name]:
                i = s if side == 0 else assign[cp.name]
                j = assign[cc.name] if side == 0 else s
                cost += float(m[i, j])
            return cost

        for _sweep in range(3):
            changed = False
            for c in self.clusters:
                cur = assign[c.name]
                costs = [local_cost(c, s) for s in range(len(c.strategies))]
                b = min(range(len(costs)), key=lambda k: (costs[k], k))
                if costs[b] + 1e-15 < costs[cur]:
                    assign[c.name] = b
                    changed = True
            if not changed:
                break
        total = sum(float(unary[c.name][assign[c.name]])
                    for c in self.clusters)
        for cp, cc, v, nb in self._edges():
            su = cp.strategies[assign[cp.name]]
            pv = cc.strategies[assign[cc.name]].in_placements.get(v)
            if pv is not None:
                total += reshard_cost(su.out_placements.get(v, R), pv,
                                      var_bytes.get(v, 0), self.mesh_size)
        for cp, cc, m in self._follow_edges():
            total += float(m[assign[cp.name], assign[cc.name]])
        return assign, total


def solve_mesh_dim(graph: MetaGraph, mesh_size: int, already_sharded=None,
                   output_constraints=None) -> Dict[str, ClusterStrategy]:
    clusters = graph.coarsen(mdconfig.coarsen_level
                             if mdconfig.enable_graph_coarsen else 0)
    solver = AutoFlowSolver1D(graph, mesh_size, already_sharded,
                              output_constraints)
    solver.add_coarsen_graph(clusters)
    if mdconfig.solver_mode == "beam":
        return solver.beam_search()
    return solver.ilp_solve()
