"""lower_gemm: linear-layer matmuls rewritten to the MFMA GEMM ops.

CPU-executable: the custom ops' CPU impls are the same aten math, so the
lowered graph must stay golden vs vanilla eager (the GPU kernels are
covered by tests/test_gpu_kernels.py).
"""
import copy
import os

import pytest
import torch
import torch.nn as nn


def _init_pg():
    import torch.distributed as dist
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29551")
        dist.init_process_group("gloo", rank=0, world_size=1)


class MLP(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(128, 512)
        self.fc2 = nn.Linear(512, 128)

    def forward(self, x):
        return self.fc2(torch.nn.functional.gelu(self.fc1(x)))


def test_lower_gemm_node_mix_and_golden():
    _init_pg()
    from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])

    torch.manual_seed(0)
    model = MLP()
    ref = copy.deepcopy(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    opt_ref = torch.optim.SGD(ref.parameters(), lr=0.05)

    def train_step(model, opt, x, y):
        opt.zero_grad()
        with torch.autocast("cpu", dtype=torch.bfloat16):
            loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        return loss

    compiled = easydist_compile(train_step, parallel_mode="auto")
    x = torch.randn(64, 128)
    y = torch.randn(64, 128)
    for _ in range(3):
        loss = compiled(model, opt, x, y)
        ref_loss = train_step(ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref_loss)) < 1e-3

    gm = list(compiled.compiled.values())[0].gm
    names = [getattr(n.target, "__name__", "") for n in gm.graph.nodes
             if n.op == "call_function"]
    n_nt = sum(1 for s in names
               if s in ("gemm_nt.default", "gemm_nt_act.default",
                        "gemm_nt_gelu.default", "gemm_nn.default",
                        "gemm_nn_act.default"))
    n_tn = sum(1 for s in names
               if s in ("gemm_tn.default", "gemm_tn_asum.default"))
    # 2 fwd NT + 2 dX NT (weight-transposed) and 2 dW TN
    assert n_nt >= 3, names
    assert n_tn >= 1, names
    # the gelu backward is fused into the dX GEMM epilogue (the dX is
    # an NN-layout GEMM since the gemm_nn lowering)
    assert "gemm_nn_act.default" in names or "gemm_nt_act.default" in names, \
        names
    assert "gelu_backward.default" not in names, names
    # ... and the forward gelu into the fwd GEMM epilogue
    assert "gemm_nt_gelu.default" in names, names
    assert "gelu.default" not in names, names
    # bias grads fused into the dW TN GEMMs
    assert "gemm_tn_asum.default" in names, names
    # no stray aten mm of the linear shapes left
    assert "addmm.default" not in names

    for (n1, p1), (n2, p2) in zip(model.named_parameters(),
                                  ref.named_parameters()):
        # bf16 autocast: the lowered ops use the same math but different
        # operand layouts, so grads differ at bf16 rounding level
        assert torch.allclose(p1, p2, rtol=1e-2, atol=1e-3), \
            (n1, float((p1 - p2).abs().max()))


def test_gemm_nn_op_numerics():
    # NN-layout op (dX backward): C = a @ b (+bias), CPU reference path
    torch.manual_seed(3)
    a = torch.randn(32, 64)
    b = torch.randn(64, 48)
    bias = torch.randn(48)
    out = torch.ops.easydist_amd.gemm_nn(a, b, bias)
    assert torch.allclose(out, torch.addmm(bias, a, b), atol=1e-5)
    out2 = torch.ops.easydist_amd.gemm_nn_act(a, b, None, 1, None)
    ref = torch.nn.functional.gelu(a @ b, approximate="tanh")
    assert torch.allclose(out2, ref, atol=1e-5)
