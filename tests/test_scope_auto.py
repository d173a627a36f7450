"""scope_auto: marked regions become call_module subgraphs with
unchanged numerics (reference tests/test_scope_auto/)."""
import torch
from torch.fx.experimental.proxy_tensor import make_fx

from easydist_amd.compiler.scope_auto import build_scope_modules, scope_marker


def test_scope_extraction():
    @scope_marker("mid")
    def middle(x):
        return torch.relu(x) * 2

    def f(x):
        a = x + 1
        b = middle(a)
        return b - 3

    x = torch.randn(4, 4)
    want = f(x)
    gm = make_fx(f, tracing_mode="fake")(x)
    gm = build_scope_modules(gm)
    mods = [n for n in gm.graph.nodes if n.op == "call_module"]
    assert len(mods) == 1, gm.graph
    assert "scope_mid" in mods[0].target
    got = gm(x)
    got = got[0] if isinstance(got, (tuple, list)) else got
    assert torch.allclose(got, want, rtol=1e-5, atol=1e-6)
