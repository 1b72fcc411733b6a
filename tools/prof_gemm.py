#!/usr/bin/env python3
"""Run a single own-GEMM shape in a loop for rocprofv3 PMC collection.

Usage: rocprofv3 --pmc ... -- python tools/prof_gemm.py --shape fc1 --impl own
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SHAPES = {
    "fc1": (24576, 4096, 16384),    # M, K, N (fwd)
    "fc2": (24576, 16384, 4096),
    "qkv": (24576, 4096, 12288),
    "sq8k": (8192, 8192, 8192),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--shape", default="fc1")
    ap.add_argument("--impl", default="own", choices=["own", "lt"])
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    M, K, N = SHAPES[args.shape]
    from paddle_amd import _ext
    C = _ext.get_ext()
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    wt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.02
    w = wt.t().contiguous()
    flops = 2.0 * M * K * N
    fn = (lambda: C.gemm_bf16_ex(x, wt, 0)) if args.impl == "own" \
        else (lambda: torch.matmul(x, w))
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    print(f"{args.shape} {args.impl}: {flops/dt/1e12:.1f} TF ({dt*1e3:.2f} ms)")


if __name__ == "__main__":
    main()
