"""Toy diffusion UNet through auto-SPMD (conv + transpose-conv + skip
cats + timestep conditioning), golden-tested against eager at ws2
(mirrors the reference's stable-diffusion example capability,
examples/torch/stable_diffusion.py, on synthetic data)."""
import copy

import pytest
import torch

from easydist_amd.utils.testing import spawn


def _body(world_size):
    import torch.distributed as dist

    from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
    from easydist_amd.models.unet import ToyUNet, ddpm_train_step

    easydist_setup(backend="torch", device="cpu")
    set_device_mesh(list(range(world_size)), ["spmd0"])
    torch.manual_seed(0)
    model = ToyUNet(cin=2, base=8, tdim=16)
    ref = copy.deepcopy(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, foreach=False)
    opt_ref = torch.optim.Adam(ref.parameters(), lr=1e-3, foreach=False)
    compiled = easydist_compile(ddpm_train_step)

    g = torch.Generator().manual_seed(7)
    for _ in range(2):
        x0 = torch.randn(4, 2, 16, 16, generator=g)
        t = torch.randint(0, 100, (4,), generator=g)
        noise = torch.randn(4, 2, 16, 16, generator=g)
        abar = torch.rand(4, generator=g) * 0.9 + 0.05
        loss = compiled(model, opt, x0, t, noise, abar)
        rl = ddpm_train_step(ref, opt_ref, x0, t, noise, abar)
        assert abs(float(loss) - float(rl)) < 1e-4, (float(loss), float(rl))


@pytest.mark.world2
def test_unet_auto_ws2():
    spawn(_body, args=(2,), world_size=2, port=29568)
