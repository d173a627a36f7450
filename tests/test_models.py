"""Model zoo smoke + compile tests (CPU).

BASELINE config #1: "ResNet-18 train_step @easydist_compile on CPU/gloo
world_size=1" is covered here; ViT/GAT/MoE get vanilla-step smoke tests
plus compiled variants where the trace is CPU-cheap.
"""
import copy

import pytest
import torch

from easydist_amd.utils.testing import init_single_process


def test_vit_tiny_step():
    from easydist_amd.models.vit import VIT_TINY, ViT, vit_train_step
    torch.manual_seed(0)
    m = ViT(VIT_TINY)
    opt = torch.optim.Adam(m.parameters(), lr=1e-3)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    l1 = vit_train_step(m, opt, x, y)
    l2 = vit_train_step(m, opt, x, y)
    assert torch.isfinite(l1) and torch.isfinite(l2)
    assert float(l2) < float(l1)


def test_resnet18_shape_step():
    from easydist_amd.models.resnet import resnet18_shape, resnet_train_step
    torch.manual_seed(0)
    m = resnet18_shape()
    opt = torch.optim.SGD(m.parameters(), lr=1e-2, momentum=0.9)
    x = torch.randn(4, 3, 64, 64)
    y = torch.randint(0, 10, (4,))
    l1 = resnet_train_step(m, opt, x, y)
    l2 = resnet_train_step(m, opt, x, y)
    assert torch.isfinite(l1) and torch.isfinite(l2)


def test_gat_step():
    from easydist_amd.models.gat import GAT, gat_train_step
    torch.manual_seed(0)
    m = GAT(in_dim=64, hidden=32, n_classes=8)
    opt = torch.optim.Adam(m.parameters(), lr=1e-3)
    x = torch.randn(32, 64)
    adj = (torch.rand(32, 32) < 0.2).float()
    adj.fill_diagonal_(1)
    y = torch.randint(0, 8, (32,))
    l1 = gat_train_step(m, opt, x, adj, y)
    l2 = gat_train_step(m, opt, x, adj, y)
    assert torch.isfinite(l1) and torch.isfinite(l2)
    assert float(l2) < float(l1)


def test_resnet18_compiled_golden():
    """BASELINE config #1: ResNet-18 @easydist_compile, CPU, ws1."""
    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.resnet import resnet18_shape

    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])

    def step(model, opt, x, y):
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        opt.zero_grad(True)
        return loss

    torch.manual_seed(0)
    model = resnet18_shape().eval()   # eval: BN uses running stats —
    # training-mode BN buffer mutation exercises the dp path below instead
    model_ref = copy.deepcopy(model)
    opt = torch.optim.SGD(model.parameters(), lr=1e-2)
    opt_ref = torch.optim.SGD(model_ref.parameters(), lr=1e-2)
    compiled = easydist_compile(step, cuda_graph=False)
    torch.manual_seed(3)
    for i in range(2):
        x = torch.randn(4, 3, 32, 32)
        y = torch.randint(0, 10, (4,))
        loss = compiled(model, opt, x, y)
        ref = step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref)) < 1e-3, \
            (i, float(loss), float(ref))


def test_gat_compiled_golden():
    """GAT through auto-SPMD (masked softmax, dense adjacency, nan_to_num
    — value-heavy ops must not mint false rules)."""
    import copy

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.gat import GAT, gat_train_step
    from easydist_amd.utils.testing import init_single_process

    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])
    torch.manual_seed(0)
    model = GAT(in_dim=32, hidden=16, n_classes=8)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3, fused=True)
    compiled = easydist_compile(gat_train_step, cuda_graph=False)
    torch.manual_seed(3)
    x = torch.randn(24, 32)
    adj = (torch.rand(24, 24) < 0.3).float()
    adj.fill_diagonal_(1)
    y = torch.randint(0, 8, (24,))
    for i in range(2):
        loss = compiled(model, opt, x, adj, y)
        ref = gat_train_step(model_ref, opt_ref, x, adj, y)
        assert abs(float(loss) - float(ref)) < 1e-4, \
            (i, float(loss), float(ref))


def test_vit_compiled_golden():
    import copy

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.vit import ViT, ViTConfig, vit_train_step
    from easydist_amd.utils.testing import init_single_process

    init_single_process()
    easydist_setup(backend="torch", device="cpu")
    set_device_mesh([0], ["spmd0"])
    torch.manual_seed(0)
    cfg = ViTConfig(image_size=32, patch_size=8, n_layer=2, n_head=2,
                    n_embd=32, n_classes=10)
    model = ViT(cfg)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3, fused=True)
    compiled = easydist_compile(vit_train_step, cuda_graph=False)
    torch.manual_seed(3)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    for i in range(2):
        loss = compiled(model, opt, x, y)
        ref = vit_train_step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref)) < 1e-4, \
            (i, float(loss), float(ref))


def test_fused_ce_ignore_index():
    from easydist_amd.ops.ce import cross_entropy
    torch.manual_seed(0)
    lg = torch.randn(32, 11)
    tg = torch.randint(0, 11, (32,))
    tg[::5] = -100
    ref = torch.nn.functional.cross_entropy(lg, tg, ignore_index=-100)
    got = cross_entropy(lg, tg, ignore_index=-100)
    assert abs(float(ref) - float(got)) < 1e-6
    # all-ignored: defined (0), no div-by-zero
    tg_all = torch.full((32,), -100)
    assert float(cross_entropy(lg, tg_all, ignore_index=-100)) == 0.0


def _vit_ws_body(world_size):
    """ViT through auto-SPMD at world>1 (conv patch embed + attention +
    class-token cat): golden vs vanilla."""
    import copy

    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, \
        set_device_mesh
    from easydist_amd.models.vit import ViT, ViTConfig, vit_train_step

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    cfg = ViTConfig(image_size=32, patch_size=8, n_layer=2, n_head=2,
                    n_embd=32, n_classes=10)
    model = ViT(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    model_ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
    opt_ref = torch.optim.Adam(model_ref.parameters(), lr=1e-3, fused=True)
    compiled = easydist_compile(vit_train_step, cuda_graph=False)
    torch.manual_seed(3)
    for i in range(2):
        x = torch.randn(8, 3, 32, 32)
        y = torch.randint(0, 10, (8,))
        dist.broadcast(x, src=0)
        dist.broadcast(y, src=0)
        loss = compiled(model, opt, x, y)
        ref = vit_train_step(model_ref, opt_ref, x, y)
        assert abs(float(loss) - float(ref)) < 5e-3, \
            (i, float(loss), float(ref))


@pytest.mark.world2
def test_vit_auto_ws2():
    from easydist_amd.utils.testing import spawn
    spawn(_vit_ws_body, args=(2,), world_size=2, port=29666)
