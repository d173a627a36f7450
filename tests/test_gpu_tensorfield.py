"""GPU test: tensorfield cross-process memory pool over hipIpc.

Server process owns the pool; writer process allocates + registers a
buffer and fills it; reader process (separate!) looks the buffer up by
name, maps the IPC handle and reads the same bytes back.
"""
import os
import subprocess
import sys
import tempfile
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

WRITER = r"""
import sys
from easydist_amd.tensorfield import TFieldClient
c = TFieldClient(sys.argv[1])
h, off, size = c.alloc(4096)
data = bytes(range(256)) * 16
c.write_bytes(h, off, data)
c.register_param("g0", "w", h, off, len(data))
print("WRITER_OK")
"""

READER = r"""
import sys
from easydist_amd.tensorfield import TFieldClient
c = TFieldClient(sys.argv[1])
ent = c.get_param("g0", "w")
assert ent is not None, "param not registered"
h, off, size = ent
got = c.read_bytes(h, off, size)
assert got == bytes(range(256)) * 16, "IPC data mismatch"
n, b = c.stat()
assert n >= 1 and b >= 4096
print("READER_OK")
"""


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
def test_tensorfield_ipc_roundtrip(tmp_path):
    sock = str(tmp_path / "tfield.sock")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

    server = subprocess.Popen(
        [sys.executable, "-m", "easydist_amd.tensorfield.server",
         "--socket", sock], env=env, stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT, text=True)
    try:
        for _ in range(50):
            if os.path.exists(sock):
                break
            time.sleep(0.2)
        assert os.path.exists(sock), "server socket never appeared"

        w = tmp_path / "writer.py"
        w.write_text(WRITER)
        r = subprocess.run([sys.executable, str(w), sock], env=env,
                           capture_output=True, text=True, timeout=120)
        assert "WRITER_OK" in r.stdout, (r.stdout, r.stderr[-2000:])

        rd = tmp_path / "reader.py"
        rd.write_text(READER)
        r2 = subprocess.run([sys.executable, str(rd), sock], env=env,
                            capture_output=True, text=True, timeout=120)
        assert "READER_OK" in r2.stdout, (r2.stdout, r2.stderr[-2000:])
    finally:
        server.kill()
        server.wait(timeout=10)
