"""tfield-server: cross-process GPU memory pool over hipIpc.

Capability parity with reference ``easydist/torch/tensorfield/``
(server.py:55-160 IPCMemoryPool, mem_pool.py, param-group registry
server.py:16-48). MI355X re-design: no cupy — the pool talks to the HIP
runtime directly through ctypes (hipMalloc / hipIpcGetMemHandle), and
clients map buffers with hipIpcOpenMemHandle (dmabuf IPC mode,
HSA_ENABLE_IPC_MODE_LEGACY=0).

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


class Hip:
    def __init__(self):
        self.lib = ctypes.CDLL("libamdhip64.so")
        self.lib.hipMalloc.argtypes = [ctypes.POINTER(ctypes.c_void_p),
                                       ctypes.c_size_t]
        self.lib.hipFree.argtypes = [ctypes.c_void_p]
        self.lib.hipIpcGetMemHandle.argtypes = [ctypes.c_char_p,
                                                ctypes.c_void_p]
        self.lib.hipSetDevice.argtypes = [ctypes.c_int]

    def malloc(self, size: int) -> int:
        p = ctypes.c_void_p()
        rc = self.lib.hipMalloc(ctypes.byref(p), size)
        if rc != 0:
            raise MemoryError(f"hipMalloc({size}) rc={rc}")
        return p.value

    def free(self, ptr: int):
        self.lib.hipFree(ctypes.c_void_p(ptr))

    def ipc_handle(self, ptr: int) -> bytes:
        buf = ctypes.create_string_buffer(HIP_IPC_HANDLE_SIZE)
        rc = self.lib.hipIpcGetMemHandle(buf, ctypes.c_void_p(ptr))
        if rc != 0:
            raise RuntimeError(f"hipIpcGetMemHandle rc={rc}")
        return buf.raw


class IPCMemoryPool:
    """One hipMalloc per allocation, exported as an IPC handle.

    (The reference leaned on cupy's pool; allocation rate here is
    model-startup-scale, so direct hipMalloc is the simpler honest
    design. A slab layer can be added behind the same protocol.)"""

    def __init__(self, device: int = 0):
        self.hip = Hip()
        self.hip.lib.hipSetDevice(device)
        self.allocs: Dict[bytes, Tuple[int, int]] = {}   # handle -> ptr,size
        self.lock = threading.Lock()

    def alloc(self, size: int) -> Tuple[bytes, int, int]:
        ptr = self.hip.malloc(size)
        h = self.hip.ipc_handle(ptr)
        with self.lock:
            self.allocs[h] = (ptr, size)
        return h, 0, size

    def free(self, handle: bytes):
        with self.lock:
            ent = self.allocs.pop(handle, None)
        if ent:
            self.hip.free(ent[0])

    def stat(self):
        with self.lock:
            return len(self.allocs), sum(s for _, s in self.allocs.values())


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
