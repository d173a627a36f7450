"""Can torch's CUDA IPC (multiprocessing reductions) share across procs?"""
import multiprocessing as mp
import os


def child(payload, q):
    try:
        import torch
        from torch.multiprocessing.reductions import rebuild_cuda_tensor
        fn, args = payload
        t = fn(*args)
        q.put(("ok", float(t.float().sum()), t.dtype == torch.uint8))
    except Exception as e:   # noqa: BLE001
        q.put(("err", repr(e)[:300]))


def main():
    import torch
    from torch.multiprocessing.reductions import reduce_tensor
    t = torch.full((1024,), 3, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    payload = reduce_tensor(t)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=child, args=(payload, q))
    proc.start(); proc.join(120)
    while not q.empty():
        print("child:", q.get())
    print("expected sum:", 3 * 1024)


if __name__ == "__main__":
    main()
