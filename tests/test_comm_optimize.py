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
