"""CPU tests for the auxiliary components: reachability, init helpers,
runtime profiler / PerfDB."""
import torch
import torch.fx as fx
import torch.nn as nn

from easydist_amd.compiler.init_helper import (CpuModuleInitHelper,
                                               RandomInitHelper,
                                               ZeroInitHelper,
                                               materialize_module)
from easydist_amd.compiler.reachability import ReachabilityMap


def test_reachability():
    g = fx.Graph()
    a = g.placeholder("a")
    b = g.call_function(torch.relu, (a,))
    c = g.call_function(torch.neg, (a,))
    d = g.call_function(torch.add, (b, c))
    g.output((d,))
    r = ReachabilityMap(g)
    assert r.reaches(a, d)
    assert r.reaches(b, d)
    assert not r.reaches(d, a)
    assert r.concurrent(b, c)
    assert not r.concurrent(a, d)
    assert c in r.parallel_peers(b)


class _M(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc = nn.Linear(4, 4)
        self.register_buffer("buf", torch.ones(4))


def test_init_helpers_meta_materialize():
    with torch.device("meta"):
        m = _M()
    assert m.fc.weight.device.type == "meta"
    materialize_module(m, ZeroInitHelper(), device="cpu")
    assert m.fc.weight.device.type == "cpu"
    assert float(m.fc.weight.abs().sum()) == 0.0

    with torch.device("meta"):
        m2 = _M()
    cpu_twin = _M()
    materialize_module(m2, CpuModuleInitHelper(cpu_twin), device="cpu")
    assert torch.equal(m2.fc.weight, cpu_twin.fc.weight)
    assert torch.equal(m2.buf, cpu_twin.buf)

    with torch.device("meta"):
        m3 = _M()
    materialize_module(m3, RandomInitHelper(seed=1), device="cpu")
    assert m3.fc.weight.std() > 0


def test_runtime_profiler_cpu(tmp_path):
    from easydist_amd.compiler.passes.runtime_prof import (PerfDB,
                                                           RuntimeProfiler)

    def f(x):
        return torch.relu(x) @ x

    gm = fx.symbolic_trace(f)
    db = PerfDB(path=str(tmp_path / "perf.db"))
    prof = RuntimeProfiler(gm, db)
    durations = prof.profile((torch.randn(64, 64),))
    assert durations and all(v >= 0 for v in durations.values())
    # cached on second run
    db2 = PerfDB(path=str(tmp_path / "perf.db"))
    assert db2._db
