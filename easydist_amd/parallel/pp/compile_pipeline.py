"""Split the whole-step trace into per-stage fw/bw/step GraphModules.

Capability parity with reference ``easydist/torch/experimental/pp/
compile_pipeline.py`` (762-1087: split_by step_split / fw_bw_split,
per-stage CompiledStage with fw_gm/bw_gm/step subgraph). Re-designed:

* ONE trace (fwd+bwd+optimizer, decomposed optimizer math) is segmented by
  marker node POSITIONS — valid because make_fx records the forward
  sequentially and the autograd engine replays the backward in strict
  reverse order, so stage regions are contiguous in trace order;
* per-stage step subgraphs fall out of a dependency walk from each stage's
  params (the decomposed Adam math is per-parameter chains — nothing to
  mask, unlike the reference's ``_foreach_*`` list surgery);
* the in-place ``copy_`` state updates are KEPT: at runtime each stage
  mutates its own persistent state tensors, no writeback pass needed.
"""
from __future__ import annotations

import logging
import operator
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set, Tuple

import torch
import torch.fx as fx
import torch.utils._pytree as pytree

from .split import LOSS_BOUNDARY

logger = logging.getLogger(__name__)

PP_SPLIT = torch.ops.easydist_amd.pp_split.default
STEP_SPLIT = torch.ops.easydist_amd.step_split.default


@dataclass
class StageGraphs:
    """One pipeline stage's three executable graphs + interface lists.

    All interface lists are names of nodes in the ORIGINAL traced graph;
    the runtime keeps an environment {name: tensor} per microbatch.
    """
    stage_idx: int
    fw_gm: fx.GraphModule
    fw_inputs: List[str]
    fw_outputs: List[str]
    bw_gm: fx.GraphModule
    bw_inputs: List[str]
    bw_outputs: List[str]
    step_gm: Optional[fx.GraphModule]
    step_inputs: List[str]
    # functional (hybrid) mode: step outputs + state writeback mapping
    step_outputs: List[str] = field(default_factory=list)
    writeback: Dict[str, str] = field(default_factory=dict)  # ph -> src
    # boundary activations: recv from prev stage / send to next
    fw_recv: List[str] = field(default_factory=list)
    fw_send: List[str] = field(default_factory=list)
    # boundary grads: recv from next stage / send to prev
    bw_recv: List[str] = field(default_factory=list)
    bw_send: List[str] = field(default_factory=list)
    # placeholders owned by this stage (params/buffers/states), graph names
    state_names: List[str] = field(default_factory=list)
    # data placeholders consumed by this stage's fw (chunked at runtime)
    data_names: List[str] = field(default_factory=list)
    # grad accumulators: bw outputs that feed step_inputs
    grad_names: List[str] = field(default_factory=list)
    # user return values produced by this stage's fw (last stage: the loss)
    ret_names: List[str] = field(default_factory=list)


@dataclass
class PipelineInfo:
    nstages: int
    stages: List[StageGraphs]
    # flat-input position -> placeholder name (traced order)
    ph_names: List[str]
    # name -> owning stage for every state placeholder
    state_stage: Dict[str, int]
    # flat OUTPUT structure: per position, the producing node name (or None)
    out_names: List[Optional[str]]
    out_spec: object
    n_state: int
    n_params: int
    param_names: List[str]       # torch param qualnames, traced order
    meta_vals: Dict[str, object]  # node name -> fake val (shape/dtype)
    # placeholder name -> user-facing qualified name (params: module
    # qualname; buffers likewise; optimizer states: "<param>.<state_key>")
    ph_qualnames: Dict[str, str] = field(default_factory=dict)
    # hybrid pp x spmd: per-placeholder / per-ret spmd placement vectors
    ph_placements: Dict[str, list] = field(default_factory=dict)
    ret_placements: Dict[str, list] = field(default_factory=dict)
    boundary_placements: Dict[str, list] = field(default_factory=dict)
    pp_mesh_dim: Optional[int] = None


def _val(node):
    return node.meta.get("val") if hasattr(node, "meta") else None


def _segment_graph(gm: fx.GraphModule):
    """Partition the node list into fw/bw segments per stage + step region.

    Returns (segments dict, marker nodes, placeholders, out_node)."""
    nodes = list(gm.graph.nodes)
    placeholders = [n for n in nodes if n.op == "placeholder"]
    out_node = nodes[-1]
    assert out_node.op == "output"

    fw_marks: List[fx.Node] = []     # boundary i markers, forward
    loss_mark: Optional[fx.Node] = None
    bw_marks: Dict[int, fx.Node] = {}
    step_marks: List[fx.Node] = []
    loss_twins: List[fx.Node] = []   # extra idx==-1 markers (identity)
    for n in nodes:
        if n.op == "call_function" and n.target is PP_SPLIT:
            idx, is_bwd = n.args[1], n.args[2]
            if idx == LOSS_BOUNDARY:
                # the FIRST forward loss marker ends the fwd region; any
                # further idx==-1 markers (chained fwd twins + their
                # backward mirrors) are plain identities to strip
                if not is_bwd and loss_mark is None:
                    loss_mark = n
                else:
                    loss_twins.append(n)
            elif is_bwd:
                bw_marks[idx] = n
            else:
                fw_marks.append(n)
        elif n.op == "call_function" and n.target is STEP_SPLIT:
            step_marks.append(n)

    nstages = len(fw_marks) + 1
    pos = {n: i for i, n in enumerate(nodes)}

    # forward segments: [start, marker] slices of the node list
    body_start = pos[placeholders[-1]] + 1 if placeholders else 0
    fw_segs = []
    prev = body_start
    for m in fw_marks:
        fw_segs.append(nodes[prev:pos[m]])
        prev = pos[m] + 1
    # last fw segment ends at the loss marker (or at bwd start)
    if loss_mark is None:
        raise RuntimeError("pp trace has no loss boundary: the train step "
                           "must call loss.backward() under SplitPatcher")
    fw_segs.append(nodes[prev:pos[loss_mark]])

    # backward segments: reverse stage order, delimited by bw markers
    bw_segs = [None] * nstages
    step_begin = min(pos[m] for m in step_marks) if step_marks else pos[out_node]
    prev = pos[loss_mark] + 1
    for s in range(nstages - 1, 0, -1):
        m = bw_marks.get(s - 1)
        assert m is not None, f"missing backward marker for boundary {s-1}"
        bw_segs[s] = nodes[prev:pos[m]]
        prev = pos[m] + 1
    bw_segs[0] = nodes[prev:step_begin]

    step_seg = [n for n in nodes[step_begin:pos[out_node]]
                if n.op != "output"]

    return (nstages, fw_segs, bw_segs, step_seg, fw_marks, bw_marks,
            loss_mark, loss_twins, step_marks, placeholders, out_node)


def _strip_markers(seg):
    return [n for n in seg
            if not (n.op == "call_function"
                    and n.target in (PP_SPLIT, STEP_SPLIT))]


def _marker_env(fw_marks, bw_marks, loss_mark, loss_twins, step_marks):
    """Map every marker node to its input value node (identity removal)."""
    env = {}
    for m in (list(fw_marks) + list(bw_marks.values()) + list(step_marks)
              + list(loss_twins)):
        env[m] = m.args[0]
    if loss_mark is not None:
        env[loss_mark] = loss_mark.args[0]
    return env


def _resolve(n, alias):
    while n in alias:
        n = alias[n]
    return n


def _extract(gm: fx.GraphModule, seg: List[fx.Node], alias: Dict,
             wanted_outputs: List[fx.Node]) -> Tuple[fx.GraphModule,
                                                     List[str], List[str]]:
    """Copy `seg` into a standalone GraphModule.

    Inputs = values referenced but not defined in seg (placeholders in the
    new graph, named after the original node). Outputs = wanted_outputs.
    Markers must already be stripped from seg; `alias` maps marker nodes to
    their input values.
    """
    seg_set = set(seg)
    g = fx.Graph()
    env: Dict[fx.Node, fx.Node] = {}
    inputs: List[str] = []
    attrs = {}

    def lookup(n: fx.Node) -> fx.Node:
        n = _resolve(n, alias)
        if n in env:
            return env[n]
        # external value -> new placeholder
        ph = g.placeholder(n.name)
        val = _val(n)
        if val is not None:
            ph.meta["val"] = val
        env[n] = ph
        inputs.append(n.name)
        return ph

    for n in seg:
        if n.op == "get_attr":
            # copy the constant attribute onto the new module
            attrs[n.target] = gm
            env[n] = None  # placeholder; replaced below
    for n in seg:
        if n.op == "get_attr":
            new = g.get_attr(n.target)
            new.meta.update(n.meta)
            env[n] = new
            continue
        env[n] = g.node_copy(n, lookup)

    outs = []
    for o in wanted_outputs:
        o = _resolve(o, alias)
        if o in env:
            outs.append(env[o])
        else:
            outs.append(lookup(o))
    g.output(tuple(outs))

    sub = fx.GraphModule(gm, g)   # gm as root: get_attr targets resolve
    sub.graph.lint()
    sub.recompile()
    return sub, inputs, [_resolve(o, alias).name for o in wanted_outputs]


def compile_pipeline(gm: fx.GraphModule, flat_inputs, n_params: int,
                     n_state: int, param_names: List[str],
                     io_map: Optional[Dict[str, str]] = None
                     ) -> PipelineInfo:
    """io_map (functional/hybrid mode): placeholder name -> producing node
    name for state round-trips — the canonicalized graph carries no
    copy_ mutations, so each stage writes its updated state back from
    its step outputs."""
    (nstages, fw_segs, bw_segs, step_seg, fw_marks, bw_marks, loss_mark,
     loss_twins, step_marks, placeholders, out_node) = _segment_graph(gm)

    alias = _marker_env(fw_marks, bw_marks, loss_mark, loss_twins,
                        step_marks)
    fw_segs = [_strip_markers(s) for s in fw_segs]
    bw_segs = [_strip_markers(s) for s in bw_segs]
    step_seg = _strip_markers(step_seg)

    flat_outs, out_spec = pytree.tree_flatten(out_node.args[0])
    ret_start = n_state + n_params

    # ---------------- stage assignment -------------------------------------
    node_stage: Dict[fx.Node, int] = {}
    for s, seg in enumerate(fw_segs):
        for n in seg:
            node_stage[n] = s
    for s, seg in enumerate(bw_segs):
        for n in seg:
            node_stage[n] = s

    # placeholders: params/buffers by fw use; states via step deps; data by
    # fw use. A placeholder used by several stages is an error for state,
    # fine for data (sent to each stage that needs it).
    ph_stage: Dict[fx.Node, Set[int]] = {p: set() for p in placeholders}
    for n in node_stage:
        for inp in n.all_input_nodes:
            inp = _resolve(inp, alias)
            if inp.op == "placeholder":
                ph_stage[inp].add(node_stage[n])

    # step-region ownership: bidirectional stage propagation to fixpoint.
    # Forward: a node inherits the stages of its inputs (param/state
    # placeholders, grads from the bwd region, other step nodes).
    # Backward: a node inherits its step-region USERS' stages — this binds
    # the per-parameter step-counter chains (add_ -> pow -> ... ->
    # addcdiv_) whose only placeholder inputs are states not yet assigned.
    step_set = set(step_seg)
    step_node_stages: Dict[fx.Node, Set[int]] = {n: set() for n in step_seg}
    for _ in range(len(step_seg)):
        changed = False
        for n in step_seg:
            stages = set(step_node_stages[n])
            for inp in n.all_input_nodes:
                inp = _resolve(inp, alias)
                if inp.op == "placeholder":
                    if len(ph_stage[inp]) == 1:
                        stages |= ph_stage[inp]
                elif inp in step_set:
                    stages |= step_node_stages[inp]
                elif inp in node_stage:          # grads from bwd
                    stages.add(node_stage[inp])
            if stages != step_node_stages[n]:
                step_node_stages[n] = stages
                changed = True
        for n in reversed(step_seg):
            stages = set(step_node_stages[n])
            for u in n.users:
                if u in step_set:
                    stages |= step_node_stages[u]
            if stages != step_node_stages[n]:
                step_node_stages[n] = stages
                changed = True
        # bind state placeholders used only inside the step region
        for n in step_seg:
            for inp in n.all_input_nodes:
                inp = _resolve(inp, alias)
                if inp.op == "placeholder" and not ph_stage[inp]:
                    ph_stage[inp] |= step_node_stages[n]
        if not changed:
            break
    shared_inplace = [n.name for n in step_seg
                      if len(step_node_stages[n]) > 1
                      and n.op == "call_function"
                      and getattr(n.target, "_schema", None) is not None
                      and n.target._schema.is_mutable]
    if shared_inplace:
        raise RuntimeError(
            f"optimizer ops shared across pipeline stages: {shared_inplace}")

    multi = [p.name for p, ss in ph_stage.items()
             if len(ss) > 1 and placeholders.index(p) < n_state]
    if multi:
        raise RuntimeError(
            f"parameters/buffers used by multiple pipeline stages "
            f"(tied weights across a split): {multi}")

    state_stage: Dict[str, int] = {}
    for i, p in enumerate(placeholders):
        if i < n_state:
            ss = ph_stage[p]
            state_stage[p.name] = next(iter(ss)) if ss else 0

    # ---------------- interface computation --------------------------------
    # users outside a segment determine its outputs
    def seg_outputs(seg, later_consumers) -> List[fx.Node]:
        seg_set = set(seg)
        outs, seen = [], set()
        for n in seg:
            for u in n.users:
                u2 = _resolve(u, alias)
                if u2 not in seg_set and n not in seen:
                    if u2 in later_consumers or u2 is out_node \
                            or u2 in alias.values():
                        outs.append(n)
                        seen.add(n)
                        break
                if u is out_node and n not in seen:
                    outs.append(n)
                    seen.add(n)
                    break
        return outs

    all_nodes_set = set(node_stage) | step_set | {out_node}

    stages: List[StageGraphs] = []
    meta_vals: Dict[str, object] = {}
    for n in gm.graph.nodes:
        v = _val(n)
        if v is not None:
            meta_vals[n.name] = v

    # boundary values (post-alias resolution)
    fw_boundary = [_resolve(m, alias) for m in fw_marks]       # idx i: fw i->i+1
    bw_boundary = {i: _resolve(m, alias) for i, m in bw_marks.items()}
    loss_val = _resolve(loss_mark, alias)

    ret_nodes = {o for o in flat_outs[ret_start:] if isinstance(o, fx.Node)}

    for s in range(nstages):
        fw_seg, bw_seg = fw_segs[s], bw_segs[s]
        fw_seg_set, bw_seg_set = set(fw_seg), set(bw_seg)

        # fw outputs: values used outside the fw segment
        fw_outs: List[fx.Node] = []
        for n in fw_seg:
            used_outside = False
            for u in n.users:
                if u.op == "call_function" and u.target in (PP_SPLIT,
                                                            STEP_SPLIT):
                    if any(uu not in fw_seg_set for uu in u.users) \
                            or u is loss_mark or u in fw_marks:
                        used_outside = True
                elif u not in fw_seg_set:
                    used_outside = True
            if used_outside and n not in fw_outs:
                fw_outs.append(n)

        bw_outs: List[fx.Node] = []
        for n in bw_seg:
            for u in n.users:
                if u.op == "call_function" and u.target in (PP_SPLIT,
                                                            STEP_SPLIT):
                    if any(uu not in bw_seg_set for uu in u.users):
                        bw_outs.append(n)
                        break
                elif u not in bw_seg_set:
                    bw_outs.append(n)
                    break

        fw_gm, fw_in, fw_out_names = _extract(gm, fw_seg, alias, fw_outs)
        bw_gm, bw_in, bw_out_names = _extract(gm, bw_seg, alias, bw_outs)

        # stage-s nodes; ∅-stage nodes are dead (no path to any parameter)
        my_step = [n for n in step_seg if s in step_node_stages[n]]
        step_gm, step_in = None, []
        step_out_names: List[str] = []
        writeback: Dict[str, str] = {}
        if my_step:
            wanted_step: List[fx.Node] = []
            if io_map is not None:
                # functional mode: the state round trip is positional —
                # output i of (params, buffers, states, ...) is the new
                # value of placeholder i (io_map only covers copy_-style
                # traces; in-place-op traces have an empty io_map)
                my_names = {n.name for n in my_step}
                for i in range(min(n_state, len(flat_outs),
                                   len(placeholders))):
                    o = flat_outs[i]
                    if isinstance(o, fx.Node) and o.op != "placeholder" \
                            and o.name in my_names:
                        writeback[placeholders[i].name] = o.name
                for ph_name, src_name in io_map.items():
                    if src_name in my_names:
                        writeback[ph_name] = src_name
                wanted_step = [n for n in my_step
                               if n.name in set(writeback.values())]
            step_gm, step_in, step_out_names = _extract(gm, my_step, alias,
                                                        wanted_step)

        sg = StageGraphs(
            stage_idx=s, fw_gm=fw_gm, fw_inputs=fw_in,
            fw_outputs=fw_out_names, bw_gm=bw_gm, bw_inputs=bw_in,
            bw_outputs=bw_out_names, step_gm=step_gm, step_inputs=step_in,
            step_outputs=step_out_names, writeback=writeback)

        # classify interfaces by name
        ph_by_name = {p.name: i for i, p in enumerate(placeholders)}
        if s > 0:
            sg.fw_recv = [fw_boundary[s - 1].name]
        if s < nstages - 1:
            sg.fw_send = [fw_boundary[s].name]
            sg.bw_recv = [bw_boundary[s].name]
        if s > 0:
            sg.bw_send = [bw_boundary[s - 1].name]
        for name in fw_in:
            if name in ph_by_name:
                if ph_by_name[name] < n_state:
                    sg.state_names.append(name)
                else:
                    sg.data_names.append(name)
        for name in bw_in + sg.step_inputs:
            if name in ph_by_name and ph_by_name[name] < n_state \
                    and name not in sg.state_names:
                sg.state_names.append(name)
        # grads: bw outputs consumed by the step region
        step_input_names = set(sg.step_inputs)
        sg.grad_names = [n for n in bw_out_names if n in step_input_names]
        # user returns produced by this stage
        sg.ret_names = [n.name for n in fw_seg if n in ret_nodes]
        if loss_val.name in fw_out_names or any(
                n.name == loss_val.name for n in fw_seg):
            if loss_val.name not in sg.ret_names \
                    and loss_val in ret_nodes:
                sg.ret_names.append(loss_val.name)
        stages.append(sg)

    out_names = [o.name if isinstance(o, fx.Node) else None
                 for o in flat_outs]

    return PipelineInfo(
        nstages=nstages, stages=stages,
        ph_names=[p.name for p in placeholders],
        state_stage=state_stage, out_names=out_names, out_spec=out_spec,
        n_state=n_state, n_params=n_params, param_names=param_names,
        meta_vals=meta_vals)
