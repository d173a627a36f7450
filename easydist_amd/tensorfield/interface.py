"""TFieldClient: python client of the tfield memory-pool server.

Capability parity with reference ``easydist/torch/tensorfield/
interface.py`` (TFieldClient:18, param-group APIs 56-120). Buffers are
mapped via torch's CUDA IPC reductions (the dmabuf path this host
driver supports); a "handle" on the wire is 8 key bytes + the pickled
``reduce_tensor`` payload.
"""
from __future__ import annotations

import socket
from typing import Optional, Tuple


class TFieldClient:
    def __init__(self, path: str = "/tmp/easydist_tfield.sock"):
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.connect(path)
        self.f = self.sock.makefile("rw")
        import os
        self._cmd(f"hello {os.getpid()}")

    def _cmd(self, line: str) -> str:
        self.f.write(line + "\n")
        self.f.flush()
        return self.f.readline().strip()

    def alloc(self, size: int) -> Tuple[bytes, int, int]:
        rep = self._cmd(f"alloc {size}")
        h, off, sz = rep.split()
        return bytes.fromhex(h), int(off), int(sz)

    def free(self, handle: bytes):
        assert self._cmd(f"free {handle.hex()}") == "ok"

    def register_param(self, group: str, name: str, handle: bytes,
                       offset: int, size: int):
        assert self._cmd(
            f"reg {group} {name} {handle.hex()} {offset} {size}") == "ok"

    def get_param(self, group: str, name: str
                  ) -> Optional[Tuple[bytes, int, int]]:
        rep = self._cmd(f"get {group} {name}")
        if rep == "none":
            return None
        h, off, sz = rep.split()
        return bytes.fromhex(h), int(off), int(sz)

    def list_params(self, group: str):
        rep = self._cmd(f"list {group}")
        return [s for s in rep.split(",") if s]

    def stat(self):
        n, b = self._cmd("stat").split()
        return int(n), int(b)

    def shutdown_server(self):
        self._cmd("quit")

    # ----------------------------------------------------- data helpers ----
    def map_tensor(self, handle: bytes):
        """Rebuild the shared CUDA tensor in THIS process."""
        import pickle
        fn, args = pickle.loads(handle[8:])
        return fn(*args)

    def write_bytes(self, handle: bytes, offset: int, data: bytes):
        import torch
        t = self.map_tensor(handle)
        src = torch.frombuffer(bytearray(data), dtype=torch.uint8)
        t[offset:offset + len(data)].copy_(src.to(t.device))
        torch.cuda.synchronize()

    def read_bytes(self, handle: bytes, offset: int, size: int) -> bytes:
        t = self.map_tensor(handle)
        return bytes(t[offset:offset + size].cpu().numpy().tobytes())
