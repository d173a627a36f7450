"""CPU tests for the comm-overlap pass: starts raised to producers, waits
sunk to first consumer, graph semantics preserved (gloo ws2 golden runs
through compile_auto with the pass enabled — tests/test_spmd_e2e.py)."""
import torch
import torch.fx as fx

from easydist_amd.compiler.passes.comm_optimize import (
    sink_waits_raise_starts)
from easydist_amd.runtime import comm_runtime as crt


def _toy_graph():
    """a -> start -> wait -> consumer, with unrelated compute between
    producer and start and between wait and consumer."""
    g = fx.Graph()
    a = g.placeholder("a")
    b = g.placeholder("b")
    prod = g.call_function(torch.add, (a, b))
    x1 = g.call_function(torch.relu, (b,))
    x2 = g.call_function(torch.relu, (x1,))
    start = g.call_function(crt.rt_all_reduce_start, (prod, "sum", 0))
    wait = g.call_function(crt.rt_wait, (start,))
    x3 = g.call_function(torch.relu, (x2,))
    out = g.call_function(torch.add, (wait, x3))
    g.output((out,))
    gm = fx.GraphModule(torch.nn.Module(), g)
    return gm


def test_sink_and_raise():
    gm = _toy_graph()
    moved = sink_waits_raise_starts(gm)
    assert moved >= 2
    nodes = list(gm.graph.nodes)
    names = [n.name for n in nodes]
    idx = {n.name: i for i, n in enumerate(nodes)}
    # start directly after its producer (prod)
    start = next(n for n in nodes if n.target is crt.rt_all_reduce_start)
    prod = start.all_input_nodes[0]
    assert idx[start.name] == idx[prod.name] + 1, names
    # wait directly before its first consumer (the final add)
    wait = next(n for n in nodes if n.target is crt.rt_wait)
    consumer = next(iter(wait.users))
    assert idx[wait.name] == idx[consumer.name] - 1, names
    # the x1/x2/x3 compute now lives inside the start..wait window
    assert idx[start.name] < idx["relu"] or idx[start.name] < idx["relu_2"]


def test_comm_cse_dedupes_identical_collectives():
    import torch.fx as fx

    from easydist_amd.compiler.passes.comm_optimize import comm_cse
    from easydist_amd.runtime import comm_runtime as crt

    g = fx.Graph()
    x = g.placeholder("x")
    s1 = g.call_function(crt.rt_all_gather_start, (x, 0, 0))
    w1 = g.call_function(crt.rt_wait, (s1,))
    a = g.call_function(torch.relu, (w1,))
    s2 = g.call_function(crt.rt_all_gather_start, (x, 0, 0))  # duplicate
    w2 = g.call_function(crt.rt_wait, (s2,))
    b = g.call_function(torch.tanh, (w2,))
    # different args: NOT a duplicate
    s3 = g.call_function(crt.rt_all_gather_start, (x, 1, 0))
    w3 = g.call_function(crt.rt_wait, (s3,))
    g.output((a, b, w3))
    gm = fx.GraphModule(torch.nn.Module(), g)
    removed = comm_cse(gm)
    assert removed == 1
    starts = [n for n in gm.graph.nodes
              if n.target is crt.rt_all_gather_start]
    assert len(starts) == 2
    # tanh now consumes the FIRST wait's result
    tanh = next(n for n in gm.graph.nodes if n.target is torch.tanh)
    assert tanh.args[0].args[0] is starts[0]


def test_milp_schedule_overlaps_and_stays_topological():
    """Exact RCPSP: with one transfer and independent compute, the MILP
    must start the transfer at t=0 (full overlap) and return a valid
    topological order with the dependent consumer after the wait."""
    from easydist_amd.schedule.rcpsp import milp_schedule

    gm = _toy_graph()
    durations = {"add": 1.0, "relu": 2.0, "relu_1": 2.0, "relu_2": 2.0,
                 "rt_all_reduce_start": 5.0}
    order = milp_schedule(gm, durations)
    assert order is not None
    idx = {n.name: i for i, n in enumerate(order)}
    # topological validity
    for i, n in enumerate(order):
        for p in n.all_input_nodes:
            assert idx[p.name] < i, (p.name, n.name)
    # the transfer overlaps compute: start comes before at least two of
    # the independent relus (sequentially it was after relu/relu_1)
    assert idx["rt_all_reduce_start"] < idx["relu_2"]
    # output node last
    assert order[-1].op == "output"


def test_milp_schedule_serializes_same_resource():
    """Two independent transfers share the comm resource: the MILP must
    NOT start both at once — makespan >= sum of transfer durations when
    compute is negligible."""
    from easydist_amd.schedule.rcpsp import milp_schedule

    g = fx.Graph()
    a = g.placeholder("a")
    b = g.placeholder("b")
    s1 = g.call_function(crt.rt_all_reduce_start, (a, "sum", 0))
    w1 = g.call_function(crt.rt_wait, (s1,))
    s2 = g.call_function(crt.rt_all_reduce_start, (b, "sum", 0))
    w2 = g.call_function(crt.rt_wait, (s2,))
    out = g.call_function(torch.add, (w1, w2))
    g.output((out,))
    gm = fx.GraphModule(torch.nn.Module(), g)
    order = milp_schedule(gm, {"rt_all_reduce_start": 3.0,
                               "rt_all_reduce_start_1": 3.0,
                               "add": 0.1})
    assert order is not None
    idx = {n.name: i for i, n in enumerate(order)}
    for i, n in enumerate(order):
        for p in n.all_input_nodes:
            assert idx[p.name] < i
