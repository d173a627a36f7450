"""GPU test of the static memory plan end-to-end (PROFILE -> pack ->
RUNTIME arena playback). Runs in a subprocess because the pluggable
allocator must be installed before the first device allocation."""
import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

SCRIPT = r"""
import json, os, torch
os.environ["EASYDIST_MEM_OPT"] = "1"
os.environ["MASTER_ADDR"] = "127.0.0.1"
os.environ["MASTER_PORT"] = "29613"   # parent pytest holds the default port
import easydist_amd.config as cfg
cfg.enable_memory_opt = True
from easydist_amd import easydist_compile, easydist_setup, set_device_mesh
from easydist_amd.utils.testing import init_single_process
import easydist_amd.memory.meta_allocator as ma

init_single_process()
easydist_setup(backend="torch", device="cuda")
assert ma.allocator_installed(), "pluggable allocator must install"
set_device_mesh([0], ["spmd0"])

import torch.nn as nn
torch.manual_seed(0)
model = nn.Sequential(nn.Linear(256, 512), nn.ReLU(),
                      nn.Linear(512, 512), nn.ReLU(),
                      nn.Linear(512, 256)).cuda()
opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)

def train_step(model, opt, x, y):
    loss = ((model(x) - y) ** 2).mean()
    loss.backward(); opt.step(); opt.zero_grad(True)
    return loss

compiled = easydist_compile(train_step, memory_opt=True, cuda_graph=False)
x = torch.randn(64, 256, device="cuda")
y = torch.randn(64, 256, device="cuda")
losses = [float(compiled(model, opt, x, y)) for _ in range(6)]
rt = list(compiled.compiled.values())[0]
stats = rt._mem_plan
assert stats is not None and stats["arena_bytes"] > 0
assert losses[-1] < losses[0], losses
print("MEMPLAN_OK", json.dumps({"losses": losses, **stats}))
"""


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
def test_memory_plan_end_to_end(tmp_path):
    p = tmp_path / "memplan_run.py"
    p.write_text(SCRIPT)
    env = dict(os.environ)
    env["EASYDIST_MEM_OPT"] = "1"
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run([sys.executable, str(p)], capture_output=True,
                       text=True, timeout=420, env=env)
    assert "MEMPLAN_OK" in r.stdout, (r.stdout[-3000:], r.stderr[-3000:])
    line = [l for l in r.stdout.splitlines() if "MEMPLAN_OK" in l][0]
    stats = json.loads(line.split("MEMPLAN_OK ")[1])
    # the plan must beat naive sum-of-allocations
    assert stats["arena_bytes"] < stats["naive_sum_bytes"]
