#!/usr/bin/env python3
"""Flash attention kernel timing on the GPT-2 bench shape."""
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


def main():
    ext = ops.load_extension()
    assert ext is not None
    out = []
    for (B, H, S, D) in [(64, 12, 1024, 64), (16, 16, 2048, 128)]:
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn_like(q)
        v = torch.randn_like(q)
        o, lse = ext.flash_attn_fwd(q, k, v, True)
        g = torch.randn_like(o)
        t_fwd = time_ms(lambda: ext.flash_attn_fwd(q, k, v, True))
        t_bwd = time_ms(lambda: ext.flash_attn_bwd(g, q, k, v, o, lse, True))
        # causal flops
        fl_fwd = 2.0 * B * H * S * S * D * 2 / 2
        fl_bwd = fl_fwd * 2.5
        r = {"shape": f"B{B} H{H} S{S} D{D}", "fwd_ms": t_fwd,
             "bwd_ms": t_bwd, "fwd_tf": fl_fwd / t_fwd / 1e9,
             "bwd_tf": fl_bwd / t_bwd / 1e9}
        out.append(r)
        print(json.dumps(r))


if __name__ == "__main__":
    main()
