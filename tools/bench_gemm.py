#!/usr/bin/env python3
"""GEMM shape microbench for the GPT-3-6.7B step (b8, s2048 -> M=16384).

Measures torch.matmul (rocBLAS/hipBLASLt) TF for the fwd/dgrad/wgrad
shapes, optionally under TunableOp tuning, to decide whether a
hand-written MFMA GEMM is worth it (guide: 8-phase 256^2 = 1563-1728 TF).
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SHAPES = [
    # (name, M, K, N)
    ("qkv_fwd", 16384, 4096, 12288),
    ("proj_fwd", 16384, 4096, 4096),
    ("fc1_fwd", 16384, 4096, 16384),
    ("fc2_fwd", 16384, 16384, 4096),
    ("lmhead_fwd", 16384, 4096, 50304),
]


def bench(fn, iters=10, warmup=3):
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
    ap.add_argument("--tunableop", action="store_true")
    ap.add_argument("--duration", type=int, default=100)
    ap.add_argument("--custom", action="store_true")
    args = ap.parse_args()
    if args.custom:
        bench_custom()
        return
    if args.tunableop:
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(True)
        tunable.set_max_tuning_duration(args.duration)

    for name, M, K, N in SHAPES:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
        # fwd (NN), dgrad dX = dY @ W^T (NT), wgrad dW = X^T @ dY (TN)
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * M * K * N
        t_nn = bench(lambda: torch.matmul(a, b))
        t_nt = bench(lambda: torch.matmul(dy, b.t()))
        t_tn = bench(lambda: torch.matmul(a.t(), dy))
        print(f"{name:12s} M{M} K{K} N{N}: NN {flops/t_nn/1e12:7.1f} TF  "
              f"NT {flops/t_nt/1e12:7.1f} TF  TN {flops/t_tn/1e12:7.1f} TF")
        del a, b, dy
        torch.cuda.empty_cache()


def bench_custom():
    from paddle_amd import _ext
    C = _ext.get_ext()
    print("== hand-written MFMA gemm_bf16 ==")
    for name, M, K, N in SHAPES:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * M * K * N
        # correctness (small slice check vs torch)
        out_nt = C.gemm_bf16(a, bt, True)
        ref_nt = a.float() @ bt.float().t()
        err = (out_nt.float() - ref_nt).abs().max() / ref_nt.abs().max()
        out_nn = C.gemm_bf16(a, b, False)
        ref_nn = a.float() @ b.float()
        err2 = (out_nn.float() - ref_nn).abs().max() / ref_nn.abs().max()
        t_nt = bench(lambda: C.gemm_bf16(a, bt, True))
        t_nn = bench(lambda: C.gemm_bf16(a, b, False))
        print(f"{name:12s}: NT {flops/t_nt/1e12:7.1f} TF (relerr {err:.2e})  "
              f"NN {flops/t_nn/1e12:7.1f} TF (relerr {err2:.2e})")
        del a, bt, b
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
