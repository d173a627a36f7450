"""tfield-server: cross-process GPU memory pool.

Capability parity with reference ``easydist/torch/tensorfield/``
(server.py:55-160 IPCMemoryPool, mem_pool.py, param-group registry
server.py:16-48). MI355X re-design: buffers are exported through
torch's CUDA IPC reductions — the dmabuf transport this host driver
supports (raw hipIpcOpenMemHandle returns hipErrorInvalidValue here;
see csrc/allocator_interface.cpp for the hipIpc-based C client kept for
drivers that allow it).

Run: ``python -m easydist_amd.tensorfield.server [--socket PATH]``.

Text protocol (newline-framed, space-separated):
    hello <pid>                       -> ok
    alloc <size>                      -> <handle_hex> <offset> <size>
    free <handle_hex>                 -> ok
    reg <group> <name> <handle_hex> <offset> <size>   -> ok
    get <group> <name>                -> <handle_hex> <offset> <size> | none
    list <group>                      -> <name>,<name>,...
    stat                              -> <n_allocs> <bytes>
"""
from __future__ import annotations

import argparse
import ctypes
import logging
import os
import socketserver
import threading
from typing import Dict, Tuple

logger = logging.getLogger("tfield-server")

HIP_IPC_HANDLE_SIZE = 64
DEFAULT_SOCKET = "/tmp/easydist_tfield.sock"


class IPCMemoryPool:
    """torch-allocated slabs exported via torch's CUDA IPC reductions.

    The raw hipIpcGetMemHandle/hipIpcOpenMemHandle pair is unavailable on
    this host driver (dmabuf-only IPC: get succeeds, open returns
    hipErrorInvalidValue); torch's multiprocessing reductions ride the
    supported dmabuf path, so the pool allocates torch CUDA tensors and
    ships their pickled `reduce_tensor` payloads as the "handle"."""

    def __init__(self, device: int = 0):
        import torch
        self.torch = torch
        torch.cuda.set_device(device)
        torch.cuda.init()
        self.allocs: Dict[bytes, Tuple[object, int]] = {}  # key->tensor,size
        self.lock = threading.Lock()
        self._next = 0

    def alloc(self, size: int) -> Tuple[bytes, int, int]:
        import pickle

        from torch.multiprocessing.reductions import reduce_tensor
        t = self.torch.empty(size, dtype=self.torch.uint8, device="cuda")
        payload = pickle.dumps(reduce_tensor(t))
        with self.lock:
            key = self._next.to_bytes(8, "little")
            self._next += 1
            self.allocs[key] = (t, size, payload)
        return key + payload, 0, size

    def free(self, handle: bytes):
        key = handle[:8]
        with self.lock:
            self.allocs.pop(key, None)

    def stat(self):
        with self.lock:
            return len(self.allocs), sum(e[1]
                                         for e in self.allocs.values())


class ParamGroupStore:
    """Named tensor registry so N processes share one weight copy
    (reference: server.py:16-48, interface.py:56-120)."""

    def __init__(self):
        self.groups: Dict[str, Dict[str, Tuple[bytes, int, int]]] = {}
        self.lock = threading.Lock()

    def reg(self, group, name, handle, offset, size):
        with self.lock:
            self.groups.setdefault(group, {})[name] = (handle, offset, size)

    def get(self, group, name):
        with self.lock:
            return self.groups.get(group, {}).get(name)

    def list(self, group):
        with self.lock:
            return sorted(self.groups.get(group, {}).keys())


class Handler(socketserver.StreamRequestHandler):
    def handle(self):
        pool: IPCMemoryPool = self.server.pool
        store: ParamGroupStore = self.server.store
        for raw in self.rfile:
            parts = raw.decode().strip().split()
            if not parts:
                continue
            cmd = parts[0]
            try:
                if cmd == "hello":
                    self._reply("ok")
                elif cmd == "alloc":
                    h, off, size = pool.alloc(int(parts[1]))
                    self._reply(f"{h.hex()} {off} {size}")
                elif cmd == "free":
                    pool.free(bytes.fromhex(parts[1]))
                    self._reply("ok")
                elif cmd == "reg":
                    store.reg(parts[1], parts[2], bytes.fromhex(parts[3]),
                              int(parts[4]), int(parts[5]))
                    self._reply("ok")
                elif cmd == "get":
                    ent = store.get(parts[1], parts[2])
                    self._reply("none" if ent is None else
                                f"{ent[0].hex()} {ent[1]} {ent[2]}")
                elif cmd == "list":
                    self._reply(",".join(store.list(parts[1])))
                elif cmd == "stat":
                    n, b = pool.stat()
                    self._reply(f"{n} {b}")
                elif cmd == "quit":
                    self._reply("bye")
                    self.server._stop.set()
                    return
                else:
                    self._reply(f"err unknown {cmd}")
            except Exception as e:   # noqa: BLE001
                self._reply(f"err {e}")

    def _reply(self, s: str):
        self.wfile.write((s + "\n").encode())
        self.wfile.flush()


class TFieldServer(socketserver.ThreadingUnixStreamServer):
    daemon_threads = True

    def __init__(self, path: str, device: int = 0):
        if os.path.exists(path):
            os.unlink(path)
        super().__init__(path, Handler)
        self.pool = IPCMemoryPool(device)
        self.store = ParamGroupStore()
        self._stop = threading.Event()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--socket", default=DEFAULT_SOCKET)
    ap.add_argument("--device", type=int, default=0)
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    srv = TFieldServer(args.socket, args.device)
    logger.info("tfield-server listening on %s", args.socket)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    srv._stop.wait()
    srv.shutdown()


if __name__ == "__main__":
    main()
