#!/usr/bin/env python3
"""GEMM microbenchmark: hand-written MFMA kernels vs hipBLASLt (aten).

Times every linear-layer shape of the GPT-2-small b64/s1024 bench step
(fwd NT, dX NT-with-weight-transpose, dW TN) plus square probes.
Prints TFLOP/s for both paths; this is the evidence base for the
per-shape dispatch policy (ops/gemm.py).

Run on a GPU box:  python benchmark/bench_gemm.py [--model gpt2-small]
"""
from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import easydist_amd.ops as ops


def time_ms(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    e.synchronize()
    return s.elapsed_time(e) / iters


def bench_nt(ext, M, N, K, bias=False):
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    bv = torch.randn(N, device="cuda", dtype=torch.bfloat16) if bias else None
    t_hand = time_ms(lambda: ext.gemm_nt(a, bt, bv))
    if bias:
        t_aten = time_ms(lambda: torch.addmm(bv, a, bt.t()))
    else:
        t_aten = time_ms(lambda: torch.mm(a, bt.t()))
    fl = 2.0 * M * N * K
    # numerics spot check
    c = ext.gemm_nt(a, bt, bv).float()
    ref = (torch.mm(a, bt.t()) if not bias
           else torch.addmm(bv, a, bt.t())).float()
    err = float((c - ref).abs().max())
    return {"shape": f"NT {M}x{N}x{K}{'+b' if bias else ''}",
            "hand_tf": fl / t_hand / 1e9, "aten_tf": fl / t_aten / 1e9,
            "hand_ms": t_hand, "aten_ms": t_aten, "max_err_vs_blaslt": err}


def bench_tn(ext, R, P, Q):
    a = torch.randn(R, P, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(R, Q, device="cuda", dtype=torch.bfloat16)
    t_hand = time_ms(lambda: ext.gemm_tn(a, b))
    t_aten = time_ms(lambda: torch.mm(a.t(), b))
    fl = 2.0 * R * P * Q
    c = ext.gemm_tn(a, b).float()
    ref = torch.mm(a.t(), b).float()
    err = float((c - ref).abs().max())
    return {"shape": f"TN r{R} {P}x{Q}",
            "hand_tf": fl / t_hand / 1e9, "aten_tf": fl / t_aten / 1e9,
            "hand_ms": t_hand, "aten_ms": t_aten, "max_err_vs_blaslt": err}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tokens", type=int, default=65536)  # b64 x s1024
    p.add_argument("--hidden", type=int, default=768)
    p.add_argument("--vocab", type=int, default=50304)
    p.add_argument("--only", default=None,
                   help="single shape 'nt,M,N,K' or 'tn,R,P,Q' (for PMC runs)")
    args = p.parse_args()
    assert torch.cuda.is_available()
    ext = ops.load_extension()
    assert ext is not None, "build the extension first"

    M, H, V = args.tokens, args.hidden, args.vocab
    if args.only:
        parts = args.only.split(",")
        if parts[0] == "nt":
            r = bench_nt(ext, int(parts[1]), int(parts[2]), int(parts[3]))
        else:
            r = bench_tn(ext, int(parts[1]), int(parts[2]), int(parts[3]))
        print(json.dumps(r))
        return
    rows = []
    # square probes (vs guide ladder numbers)
    for s in (4096, 8192):
        rows.append(bench_nt(ext, s, s, s))
    # fwd linears
    for (N, K, b) in [(3 * H, H, True), (H, H, True), (4 * H, H, True),
                      (H, 4 * H, True), (V, H, False)]:
        rows.append(bench_nt(ext, M, N, K, bias=b))
    # dX (NT with transposed weight)
    for (N, K) in [(H, 3 * H), (H, H), (H, 4 * H), (4 * H, H), (H, V)]:
        rows.append(bench_nt(ext, M, N, K))
    # dW (TN)
    for (P, Q) in [(3 * H, H), (H, H), (4 * H, H), (H, 4 * H), (V, H)]:
        rows.append(bench_tn(ext, M, P, Q))

    for r in rows:
        print(f"{r['shape']:>28}: hand {r['hand_tf']:7.1f} TF "
              f"({r['hand_ms']:7.3f} ms)  aten {r['aten_tf']:7.1f} TF "
              f"({r['aten_ms']:7.3f} ms)  err {r['max_err_vs_blaslt']:.3f}")
    print(json.dumps(rows))


if __name__ == "__main__":
    main()
