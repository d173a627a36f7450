"""MetaIR: framework-neutral graph IR with per-node SPMD strategy pools.

Capability parity with reference ``easydist/metashard/metair.py`` (SPMD
placements, VarSPMDStrategy, MetaNode/MetaVar/MetaGraph, strategy pools,
cone coarsening; reference lines 29-961). Re-designed, torch-only, with the
1-D-per-mesh-dim strategy representation the AutoFlow solver consumes.
"""
from __future__ import annotations

import functools
import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from .combination import CombinationFunc

logger = logging.getLogger(__name__)


# --------------------------------------------------------------- placements --
@dataclass(frozen=True)
class SPMD:
    """Single-mesh-dim placement: REPLICATE, SHARD(dim), PARTIAL(op) or
    FLAT (1-D flatten+pad shard, used by the ZeRO transforms where state
    tensors are sharded regardless of their shape)."""
    state: str            # 'R' | 'S' | 'P' | 'F'
    dim: int = -1         # tensor dim for 'S'
    reduce_op: str = ""   # 'sum' | 'max' | 'min' for 'P'
    shape: Tuple[int, ...] = ()   # global shape for 'F' (unshard needs it)

    REPLICATE = "R"
    SHARD = "S"
    PARTIAL = "P"
    FLAT = "F"

    def is_replicate(self):
        return self.state == "R"

    def is_shard(self, dim=None):
        return self.state == "S" and (dim is None or self.dim == dim)

    def is_partial(self):
        return self.state == "P"

    def is_flat_shard(self):
        return self.state == "F"

    def __repr__(self):
        if self.state == "R":
            return "R"
        if self.state == "S":
            return f"S({self.dim})"
        if self.state == "F":
            return f"F{list(self.shape)}"
        return f"P({self.reduce_op})"


R = SPMD(SPMD.REPLICATE)


def S(dim: int) -> SPMD:
    return SPMD(SPMD.SHARD, dim=dim)


def P(op: str = "sum") -> SPMD:
    return SPMD(SPMD.PARTIAL, reduce_op=op)


def F(shape) -> SPMD:
    return SPMD(SPMD.FLAT, shape=tuple(shape))


# ------------------------------------------------------------------- IR ------
@dataclass
class MetaVar:
    name: str
    shape: Tuple[int, ...]
    dtype_bytes: int = 4

    @property
    def nbytes(self) -> int:
        n = self.dtype_bytes
        for s in self.shape:
            n *= s
        return n

    def __hash__(self):
        return hash(self.name)

    def __repr__(self):
        return f"MetaVar({self.name}, {list(self.shape)})"


@dataclass
class NodeSPMDStrategy:
    """One candidate per-mesh-dim strategy for a node."""
    in_placements: List[SPMD]        # one per tensor invar
    out_placements: List[SPMD]       # one per outvar

    def __repr__(self):
        return f"<in={self.in_placements} out={self.out_placements}>"


class MetaNode:
    def __init__(self, name: str, op_name: str, invars: List[MetaVar],
                 outvars: List[Optional[MetaVar]], sharding_ann=None,
                 combination_ann=None, is_placeholder: bool = False,
                 flops: float = 0.0):
        self.name = name
        self.op_name = op_name
        self.invars = invars
        self.outvars = outvars
        self.sharding_ann = sharding_ann
        self.combination_ann = combination_ann or {}
        self.is_placeholder = is_placeholder
        # FLOPs of the op at global shapes (matmul-family filled by the
        # bridge); 0 means "memory-bound: use the byte-count proxy"
        self.flops = flops
        self._pool_cache: Dict[Tuple, List[NodeSPMDStrategy]] = {}

    # ------------------------------------------------------- strategy pool ---
    def build_strategy_pool(self, mesh_size: int,
                            already_sharded: Optional[Dict[str, Dict[int, int]]] = None
                            ) -> List[NodeSPMDStrategy]:
        """Enumerate single-mesh-dim strategies consistent with the rules.

        `already_sharded` maps var name -> {tensor_dim: total_factor} from
        previously solved mesh dims; divisibility is checked against the
        remaining per-shard size so N-D sharding stays even.
        """
        key = (mesh_size, )
        # note: already_sharded varies between solver passes, so only cache
        # when it is empty
        cacheable = not already_sharded
        if cacheable and key in self._pool_cache:
            return self._pool_cache[key]
        already_sharded = already_sharded or {}

        def dim_ok(var: Optional[MetaVar], d: int) -> bool:
            if var is None or d >= len(var.shape):
                return False
            size = var.shape[d]
            prior = already_sharded.get(var.name, {}).get(d, 1)
            return size % (prior * mesh_size) == 0

        pool: List[NodeSPMDStrategy] = []
        # the all-replicate strategy is always valid
        pool.append(NodeSPMDStrategy([R] * len(self.invars),
                                     [R] * len(self.outvars)))
        if self.is_placeholder:
            # placeholders may be pre-sharded along any divisible dim
            var = self.outvars[0]
            for d in range(len(var.shape)):
                if dim_ok(var, d):
                    pool.append(NodeSPMDStrategy([], [S(d)]))
            if cacheable:
                self._pool_cache[key] = pool
            return pool

        if self.sharding_ann is not None:
            for sid, comb in self.combination_ann.items():
                positions = self.sharding_ann.positions_of(sid)
                if not positions:
                    continue
                in_pl = [R] * len(self.invars)
                ok = True
                for (i, d) in positions:
                    if not dim_ok(self.invars[i], d):
                        ok = False
                        break
                    # halo-carrying groups need boundary exchange the plain
                    # SHARD placement doesn't express; they are kept in the
                    # annotation (conv spatial sharding) but not offered to
                    # the solver until halo reshard lands
                    if self.sharding_ann[i][d].halo != 0:
                        ok = False
                        break
                    in_pl[i] = S(d)
                if not ok:
                    continue
                out_pl = self._outs_from_comb(comb, dim_ok)
                if out_pl is None:
                    continue
                pool.append(NodeSPMDStrategy(in_pl, out_pl))
        if cacheable:
            self._pool_cache[key] = pool
        return pool

    def _outs_from_comb(self, comb, dim_ok) -> Optional[List[SPMD]]:
        combs = comb if isinstance(comb, list) else [comb]
        if len(combs) != len(self.outvars):
            # single comb for single output packed in list mismatch guard
            if len(self.outvars) == 1:
                combs = [comb]
            else:
                return None
        out_pl: List[SPMD] = []
        import operator
        for var, c in zip(self.outvars, combs):
            if c is None or var is None:
                out_pl.append(R)
                continue
            if not isinstance(c, functools.partial):
                return None
            if c.func is CombinationFunc.identity:
                out_pl.append(R)
            elif c.func is CombinationFunc.reduce_mean:
                out_pl.append(P("avg"))
            elif c.func is CombinationFunc.reduce:
                op = c.keywords.get("ops", operator.add)
                name = {operator.add: "sum"}.get(op, None)
                if name is None:
                    import torch
                    name = {torch.maximum: "max", torch.minimum: "min"}.get(op, "sum")
                out_pl.append(P(name))
            elif c.func is CombinationFunc.gather:
                d = c.keywords["dim"]
                if c.keywords.get("halowidth", 0) != 0:
                    return None  # halo gather not expressible as plain SHARD
                if not dim_ok(var, d):
                    return None
                out_pl.append(S(d))
            else:
                return None
        return out_pl

    def __repr__(self):
        return f"MetaNode({self.name}: {self.op_name})"


class MetaGraph:
    def __init__(self, name: str = "graph"):
        self.name = name
        self.nodes: List[MetaNode] = []          # topo order, non-placeholder
        self.placeholders: List[MetaNode] = []
        self.output_vars: List[str] = []         # var names returned
        self.state_io_map: Dict[str, str] = {}   # input var name -> output var name
        self.var_producer: Dict[str, Tuple[MetaNode, int]] = {}
        self.var_consumers: Dict[str, List[Tuple[MetaNode, int]]] = {}
        self.vars: Dict[str, MetaVar] = {}

    def add_node(self, node: MetaNode):
        if node.is_placeholder:
            self.placeholders.append(node)
        else:
            self.nodes.append(node)
        for k, v in enumerate(node.outvars):
            if v is not None:
                self.var_producer[v.name] = (node, k)
                self.vars[v.name] = v
        for k, v in enumerate(node.invars):
            if v is not None:
                self.var_consumers.setdefault(v.name, []).append((node, k))
                self.vars.setdefault(v.name, v)

    def all_nodes(self) -> List[MetaNode]:
        return self.placeholders + self.nodes

    def liveness(self):
        """Per-op live var sets in node order (for the memory-aware solver)."""
        last_use: Dict[str, int] = {}
        for idx, node in enumerate(self.nodes):
            for v in node.invars:
                if v is not None:
                    last_use[v.name] = idx
        for name in self.output_vars:
            last_use[name] = len(self.nodes)
        live = []
        active = set()
        for idx, node in enumerate(self.nodes):
            for v in node.outvars:
                if v is not None:
                    active.add(v.name)
            live.append(set(active))
            for v in node.invars:
                if v is not None and last_use.get(v.name, -1) <= idx:
                    active.discard(v.name)
        return live

    def coarsen(self, level: int = 1) -> List["MetaNodeCluster"]:
        """Cone clustering: merge single-consumer chains into their consumer.

        Shrinks the MILP the way the reference's cone coarsening does
        (reference metair.py:852-917): the solver then picks one strategy
        per cluster, with sync-free interiors.
        """
        consumers_count: Dict[str, int] = {}
        for node in self.nodes:
            cnt = 0
            for v in node.outvars:
                if v is None:
                    continue
                cnt += len(self.var_consumers.get(v.name, []))
                if v.name in self.output_vars:
                    cnt += 1
            consumers_count[node.name] = cnt

        def is_shrink(node: MetaNode) -> bool:
            """Cone roots sit at shrinking nodes (reference metair.py:852-917):
            a node whose outputs are smaller than its inputs ends the
            sync-free region, so resharding may happen after it."""
            out_b = sum(v.nbytes for v in node.outvars if v is not None)
            in_b = sum(v.nbytes for v in node.invars if v is not None)
            return out_b < in_b

        from .. import config as mdconfig
        cluster_of: Dict[str, MetaNodeCluster] = {}
        clusters: List[MetaNodeCluster] = []
        MAX_CLUSTER = mdconfig.coarsen_max_cluster
        # reverse topo: consumers first
        for node in reversed(self.nodes):
            target_cluster = None
            is_root = consumers_count[node.name] != 1 or is_shrink(node)
            if level >= 1 and not is_root:
                # the single consumer node
                cons = None
                for v in node.outvars:
                    if v is not None:
                        lst = self.var_consumers.get(v.name, [])
                        if lst:
                            cons = lst[0][0]
                if cons is not None and cons.name in cluster_of:
                    c = cluster_of[cons.name]
                    if len(c.nodes) < MAX_CLUSTER:
                        target_cluster = c
            if target_cluster is None:
                target_cluster = MetaNodeCluster(f"cluster_{len(clusters)}", self)
                clusters.append(target_cluster)
            target_cluster.nodes.append(node)
            cluster_of[node.name] = target_cluster
        for c in clusters:
            c.nodes.reverse()  # topo order inside cluster
        clusters.reverse()
        # placeholder clusters (one per placeholder)
        ph_clusters = []
        for ph in self.placeholders:
            c = MetaNodeCluster(f"ph_{ph.name}", self, is_placeholder=True)
            c.nodes.append(ph)
            ph_clusters.append(c)
        return ph_clusters + clusters

    def __repr__(self):
        return (f"MetaGraph({self.name}: {len(self.placeholders)} placeholders, "
                f"{len(self.nodes)} nodes)")


@dataclass
class ClusterStrategy:
    node_strategies: Dict[str, NodeSPMDStrategy]
    in_placements: Dict[str, SPMD]    # external invar name -> required placement
    out_placements: Dict[str, SPMD]   # externally visible outvar -> placement
    mem_cost: float = 0.0
    comp_cost: float = 0.0            # seconds on the CDNA4 roofline


class MetaNodeCluster:
    """A group of nodes the solver assigns one joint strategy to."""

    def __init__(self, name: str, graph: MetaGraph, is_placeholder: bool = False):
        self.name = name
        self.graph = graph
        self.nodes: List[MetaNode] = []
        self.is_placeholder = is_placeholder
        self.strategies: List[ClusterStrategy] = []

    def finalize(self, mesh_size: int, already_sharded=None):
        """Enumerate joint sync-free strategies for the cluster."""
        interior_names = {n.name for n in self.nodes}
        produced_here = {}
        for n in self.nodes:
            for v in n.outvars:
                if v is not None:
                    produced_here[v.name] = n

        root = self.nodes[-1]
        root_pool = root.build_strategy_pool(mesh_size, already_sharded)
        self.strategies = []
        seen_sig = set()
        for root_strat in root_pool:
            st = self._derive(root, root_strat, interior_names, produced_here,
                              mesh_size, already_sharded)
            if st is None:
                continue
            sig = (tuple(sorted((k, repr(v)) for k, v in st.in_placements.items())),
                   tuple(sorted((k, repr(v)) for k, v in st.out_placements.items())))
            if sig in seen_sig:
                continue
            seen_sig.add(sig)
            self.strategies.append(st)
        if not self.strategies:
            # guaranteed fallback: full replicate
            ns = {}
            for n in self.nodes:
                ns[n.name] = NodeSPMDStrategy([R] * len(n.invars),
                                              [R] * len(n.outvars))
            st = ClusterStrategy(ns, {}, {})
            self._fill_interface(st)
            self.strategies.append(st)
        for st in self.strategies:
            st.mem_cost = self._mem_cost(st, mesh_size)
            st.comp_cost = self._comp_cost(st, mesh_size)
        return self.strategies

    def _derive(self, root, root_strat, interior_names, produced_here,
                mesh_size, already_sharded) -> Optional[ClusterStrategy]:
        chosen: Dict[str, NodeSPMDStrategy] = {root.name: root_strat}
        # requirement placed on vars consumed inside the cluster
        req: Dict[str, SPMD] = {}

        def record_reqs(node, strat):
            for k, v in enumerate(node.invars):
                if v is None:
                    continue
                pl = strat.in_placements[k]
                if v.name in req and repr(req[v.name]) != repr(pl):
                    return False
                req[v.name] = pl
            return True

        if not record_reqs(root, root_strat):
            return None
        # interior nodes in reverse topo order (consumers already handled)
        for node in reversed(self.nodes[:-1]):
            pool = node.build_strategy_pool(mesh_size, already_sharded)
            # requirement on this node's outputs
            want = [req.get(v.name) if v is not None else None
                    for v in node.outvars]
            cand = None
            for s in pool:
                match = True
                for k, w in enumerate(want):
                    if w is not None and repr(s.out_placements[k]) != repr(w):
                        match = False
                        break
                if match:
                    cand = s
                    break
            if cand is None:
                return None
            chosen[node.name] = cand
            if not record_reqs(node, cand):
                return None
        st = ClusterStrategy(chosen, {}, {})
        self._fill_interface(st)
        return st

    def _fill_interface(self, st: ClusterStrategy):
        interior = {n.name for n in self.nodes}
        produced_here = set()
        for n in self.nodes:
            for v in n.outvars:
                if v is not None:
                    produced_here.add(v.name)
        for n in self.nodes:
            strat = st.node_strategies[n.name]
            for k, v in enumerate(n.invars):
                if v is None or v.name in produced_here:
                    continue
                st.in_placements[v.name] = strat.in_placements[k]
            for k, v in enumerate(n.outvars):
                if v is None:
                    continue
                consumers = self.graph.var_consumers.get(v.name, [])
                external = any(c.name not in interior for c, _ in consumers)
                if external or v.name in self.graph.output_vars or not consumers:
                    st.out_placements[v.name] = strat.out_placements[k]

    def _mem_cost(self, st: ClusterStrategy, mesh_size: int) -> float:
        total = 0.0
        for n in self.nodes:
            strat = st.node_strategies.get(n.name)
            if strat is None:
                continue
            for k, v in enumerate(n.outvars):
                if v is None:
                    continue
                pl = strat.out_placements[k]
                local = v.nbytes / (mesh_size if pl.is_shard() else 1)
                total += local
        return total

    def _comp_cost(self, st: ClusterStrategy, mesh_size: int) -> float:
        """CDNA4 roofline estimate: max(MFMA time, HBM time) per node,
        divided by mesh_size when the node's work is actually partitioned
        (any output SHARD/PARTIAL)."""
        from .. import config as mdconfig
        total = 0.0
        for n in self.nodes:
            strat = st.node_strategies.get(n.name)
            if strat is None or n.is_placeholder:
                continue
            nbytes = sum(v.nbytes for v in n.outvars if v is not None)
            nbytes += sum(v.nbytes for v in n.invars if v is not None)
            t_mem = nbytes / mdconfig.HBM_BW
            t_mfma = n.flops / mdconfig.MFMA_BF16_FLOPS
            t = max(t_mem, t_mfma)
            partitioned = any(not p.is_replicate() for p in strat.out_placements)
            total += t / mesh_size if partitioned else t
        return total

    def __repr__(self):
        return f"Cluster({self.name}: {[n.name for n in self.nodes]})"
