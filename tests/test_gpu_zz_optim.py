"""GPU numerics for the fused Adam kernel's weight-decay path (coupled
L2: grad += wd*p inside the kernel) vs the fp32 aten reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fused_adam_weight_decay_gpu():
    from easydist_amd.ops import optim  # registers the op
    torch.manual_seed(0)
    dev = "cuda"
    ps = [torch.randn(1000, device=dev), torch.randn(257, device=dev)]
    gs = [torch.randn_like(p) for p in ps]
    eas = [torch.zeros_like(p) for p in ps]
    eass = [torch.zeros_like(p) for p in ps]
    sts = [torch.zeros((), device=dev) for _ in ps]
    out = torch.ops.easydist_amd.fused_adam_step(
        ps, gs, eas, eass, sts, 1e-2, 0.9, 0.999, 0.05, 1e-8)
    ref = optim._adam_aten([p.clone() for p in ps], gs,
                           [t.clone() for t in eas],
                           [t.clone() for t in eass],
                           [t.clone() for t in sts],
                           1e-2, 0.9, 0.999, 0.05, 1e-8)
    for k in range(4):
        for a, b in zip(out[k], ref[k]):
            assert torch.allclose(a.float(), b.float(), rtol=1e-5,
                                  atol=1e-6), (k, (a - b).abs().max())


def test_fused_sgd_gpu():
    from easydist_amd.ops import optim
    torch.manual_seed(0)
    dev = "cuda"
    ps = [torch.randn(1000, device=dev), torch.randn(257, device=dev)]
    gs = [torch.randn_like(p) for p in ps]
    bs = [torch.randn_like(p) for p in ps]
    out = torch.ops.easydist_amd.fused_sgd_step(
        ps, gs, bs, 1e-2, 0.9, 0.0, 0.01, True)
    ref = optim._sgd_aten([p.clone() for p in ps], gs,
                          [b.clone() for b in bs],
                          1e-2, 0.9, 0.0, 0.01, True)
    for k in range(2):
        for a, b in zip(out[k], ref[k]):
            assert torch.allclose(a, b, rtol=1e-5, atol=1e-6), \
                (k, (a - b).abs().max())
