"""Micro-test hipIpc handle passing between two processes via ctypes."""
import ctypes
import multiprocessing as mp
import os
import sys


def child(handle_bytes, q, force_ctx):
    try:
        lib = ctypes.CDLL("libamdhip64.so")
        lib.hipSetDevice.argtypes = [ctypes.c_int]
        lib.hipSetDevice(0)
        if force_ctx:
            lib.hipFree.argtypes = [ctypes.c_void_p]
            lib.hipFree(None)           # force context creation
        lib.hipIpcOpenMemHandle.argtypes = [
            ctypes.POINTER(ctypes.c_void_p), ctypes.c_char_p, ctypes.c_uint]
        p = ctypes.c_void_p()
        rc = lib.hipIpcOpenMemHandle(ctypes.byref(p), handle_bytes, 1)
        q.put(("open", rc, p.value))
        if rc == 0:
            buf = ctypes.create_string_buffer(16)
            lib.hipMemcpy.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                      ctypes.c_size_t, ctypes.c_int]
            rc2 = lib.hipMemcpy(ctypes.addressof(buf), p, 16, 2)  # D2H
            q.put(("read", rc2, buf.raw[:8].hex()))
    except Exception as e:   # noqa: BLE001
        q.put(("err", str(e), None))


def main():
    lib = ctypes.CDLL("libamdhip64.so")
    lib.hipSetDevice.argtypes = [ctypes.c_int]
    lib.hipSetDevice(0)
    lib.hipMalloc.argtypes = [ctypes.POINTER(ctypes.c_void_p),
                              ctypes.c_size_t]
    ptr = ctypes.c_void_p()
    rc = lib.hipMalloc(ctypes.byref(ptr), 4096)
    print("hipMalloc rc", rc, hex(ptr.value or 0))
    lib.hipMemset.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_size_t]
    lib.hipMemset(ptr, 0xAB, 4096)
    lib.hipIpcGetMemHandle.argtypes = [ctypes.c_char_p, ctypes.c_void_p]
    hbuf = ctypes.create_string_buffer(64)
    rc = lib.hipIpcGetMemHandle(hbuf, ptr)
    print("hipIpcGetMemHandle rc", rc, "handle head", hbuf.raw[:16].hex())
    for force_ctx in (False, True):
        q = mp.Queue()
        proc = mp.Process(target=child, args=(hbuf.raw, q, force_ctx))
        proc.start()
        proc.join(60)
        while not q.empty():
            print("force_ctx", force_ctx, q.get())


if __name__ == "__main__":
    print("HSA_ENABLE_IPC_MODE_LEGACY =",
          os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY"))
    main()
