#!/usr/bin/env python3
"""Flash-attention kernel microbenchmark on MI355X.

Reports achieved TFLOP/s for fwd and bwd at GPT-3-6.7B shapes.
attn FLOPs (causal): fwd = 2 * 2 * B*H*S^2*D * 0.5 ; bwd = 2.5x fwd.
"""
import argparse
import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import paddle_amd  # noqa: F401,E402  (patches + ext)
from paddle_amd import _ext


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--b", type=int, default=4)
    ap.add_argument("--h", type=int, default=32)
    ap.add_argument("--s", type=int, default=2048)
    ap.add_argument("--d", type=int, default=128)
    ap.add_argument("--causal", type=int, default=1)
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    C = _ext.get_ext()
    B, H, S, D = args.b, args.h, args.s, args.d
    causal = bool(args.causal)
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    scale = 1.0 / math.sqrt(D)
    o, lse = C.flash_attn_fwd(q, k, v, None, scale, causal)
    do = torch.randn_like(o)

    fwd_flops = 4 * B * H * S * S * D * (0.5 if causal else 1.0)
    t_fwd = bench(lambda: C.flash_attn_fwd(q, k, v, None, scale, causal), args.iters)
    t_bwd = bench(lambda: C.flash_attn_bwd(do, q, k, v, o, lse, None, None, None,
                                           scale, causal), args.iters)
    print(f"shape B{B} H{H} S{S} D{D} causal={causal}")
    print(f"fwd: {t_fwd*1e3:8.3f} ms  {fwd_flops/t_fwd/1e12:7.1f} TF")
    print(f"bwd: {t_bwd*1e3:8.3f} ms  {2.5*fwd_flops/t_bwd/1e12:7.1f} TF")


if __name__ == "__main__":
    main()
