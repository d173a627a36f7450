"""CPU tests: tile_comm rewrite correctness + memory ownership checker."""
import torch
import torch.fx as fx

from easydist_amd.compiler.passes.tile_comm import tile_comm
from easydist_amd.runtime import comm_runtime as crt
from easydist_amd.runtime.race_check import MemOwnershipChecker
from easydist_amd.utils.testing import init_single_process


def test_tile_comm_rewrite_and_numerics():
    from easydist_amd import easydist_setup, set_device_mesh
    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])

    g = fx.Graph()
    a = g.placeholder("a")
    s = g.call_function(crt.rt_all_reduce_start, (a, "sum", 0))
    w = g.call_function(crt.rt_wait, (s,))
    out = g.call_function(torch.relu, (w,))
    g.output((out,))
    gm = fx.GraphModule(torch.nn.Module(), g)

    x = torch.randn(64, 1024)
    # fake meta so the pass sees the size
    ph = next(n for n in gm.graph.nodes if n.op == "placeholder")
    ph.meta["val"] = x
    want = gm(x)[0].clone()
    n = tile_comm(gm, n_tiles=4, threshold_bytes=1)
    assert n == 1
    got = gm(x)[0]
    assert torch.allclose(got, want)
    starts = [nd for nd in gm.graph.nodes
              if nd.op == "call_function"
              and nd.target is crt.rt_all_reduce_start]
    assert len(starts) == 4


def test_mem_ownership_checker_clean_graph():
    def f(x):
        a = torch.relu(x)
        return a + x

    gm = fx.symbolic_trace(f)
    checker = MemOwnershipChecker(gm)
    violations = checker.check((torch.randn(8, 8),))
    assert violations == []
