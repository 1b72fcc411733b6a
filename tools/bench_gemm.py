#!/usr/bin/env python3
"""GEMM microbench + autotune-table generation for the model hot shapes.

--autotune: times own MFMA kernel vs hipBLASLt (torch.matmul) for every
(layout, M, N, K) the flagship models hit, verifies numerics vs fp32,
and writes paddle_amd/ops/gemm_table.json -- the committed dispatch
table (reference pattern: matmul_kernel_impl.h:914-958).

Layout conventions (paddle Linear W [K, N]):
  nt fwd:   C[M,N] = X[M,K] @ Wt[N,K]^T      (Wt = cached transpose)
  nt dgrad: dX[M,K] = dY[M,N] @ W[K,N]^T'    (W is already NT's B-operand)
  tn wgrad: dW[K,N] = X[M,K]^T @ dY[M,N]
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# (name, M, K, N) -- fwd GEMM shapes; dgrad/wgrad are derived
SHAPES_B8 = [
    ("qkv", 16384, 4096, 12288),
    ("proj", 16384, 4096, 4096),
    ("fc1", 16384, 4096, 16384),
    ("fc2", 16384, 16384, 4096),
    ("lmhead", 16384, 4096, 50304),
]
SHAPES_B12 = [
    ("qkv", 24576, 4096, 12288),
    ("proj", 24576, 4096, 4096),
    ("fc1", 24576, 4096, 16384),
    ("fc2", 24576, 16384, 4096),
    ("lmhead", 24576, 4096, 50304),
]
SHAPES_BERT = [
    ("bert_qkv", 32768, 768, 2304),
    ("bert_fc1", 32768, 768, 3072),
    ("bert_fc2", 32768, 3072, 768),
]
SHAPES_LLAMA = [
    ("ll_qkv", 8192, 4096, 6144),
    ("ll_gateup", 8192, 4096, 22016),
    ("ll_down", 8192, 11008, 4096),
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
    ap.add_argument("--autotune", action="store_true")
    ap.add_argument("--fp8", action="store_true")
    ap.add_argument("--iters", type=int, default=10)
    args = ap.parse_args()
    if args.autotune:
        autotune(args.iters)
        return
    if args.fp8:
        bench_fp8()
        return
    if args.custom:
        bench_custom()
        return
    if args.tunableop:
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(True)
        tunable.set_max_tuning_duration(args.duration)

    for name, M, K, N in SHAPES_B8:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
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
    print("== hand-written MFMA gemm_bf16_ex ==")
    for name, M, K, N in SHAPES_B8:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * M * K * N
        out_nt = C.gemm_bf16_ex(a, bt, 0)[0]
        ref_nt = a.float() @ bt.float().t()
        err = (out_nt.float() - ref_nt).abs().max() / ref_nt.abs().max()
        out_nn = C.gemm_bf16_ex(a, b, 1)[0]
        ref_nn = a.float() @ b.float()
        err2 = (out_nn.float() - ref_nn).abs().max() / ref_nn.abs().max()
        t_nt = bench(lambda: C.gemm_bf16_ex(a, bt, 0))
        t_nn = bench(lambda: C.gemm_bf16_ex(a, b, 1))
        print(f"{name:12s}: NT {flops/t_nt/1e12:7.1f} TF (relerr {err:.2e})  "
              f"NN {flops/t_nn/1e12:7.1f} TF (relerr {err2:.2e})")
        del a, bt, b
        torch.cuda.empty_cache()


def _check(out, ref, tag, tol=3e-2):
    err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
    assert err < tol, f"{tag}: relerr {err:.3e}"
    return err


def autotune(iters):
    """Time own vs hipBLASLt per (layout, shape); write gemm_table.json."""
    from paddle_amd import _ext
    C = _ext.get_ext()
    entries = {}
    shapes = SHAPES_B8 + SHAPES_B12 + SHAPES_BERT + SHAPES_LLAMA
    own_wins_nt = 0
    nt_total = 0
    for name, M, K, N in shapes:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = (torch.randn(K, N, device="cuda", dtype=torch.bfloat16) * 0.02)
        wt = w.t().contiguous()
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * M * K * N

        # ---- fwd: nt (x @ wt^T) --------------------------------------------
        _check(C.gemm_bf16_ex(x, wt, 0)[0], x.float() @ w.float(), f"{name} fwd")
        t_own = bench(lambda: C.gemm_bf16_ex(x, wt, 0), iters)
        t_lt = bench(lambda: torch.matmul(x, w), iters)
        own_tf, lt_tf = flops / t_own / 1e12, flops / t_lt / 1e12
        entries[f"nt:{M}x{N}x{K}"] = {
            "impl": "own" if own_tf > lt_tf else "blaslt",
            "own_tf": round(own_tf, 1), "blaslt_tf": round(lt_tf, 1), "src": name}
        nt_total += 1
        own_wins_nt += own_tf > lt_tf
        print(f"{name:10s} fwd  nt M{M} N{N} K{K}: own {own_tf:7.1f} lt {lt_tf:7.1f}")

        # ---- dgrad: nt (dy @ w^T == nt with Bt=w[K,N]... B-operand is w) ---
        _check(C.gemm_bf16_ex(dy, w, 0)[0], dy.float() @ w.float().t(), f"{name} dgrad")
        t_own = bench(lambda: C.gemm_bf16_ex(dy, w, 0), iters)
        t_lt = bench(lambda: torch.matmul(dy, w.t()), iters)
        own_tf, lt_tf = flops / t_own / 1e12, flops / t_lt / 1e12
        entries[f"nt:{M}x{K}x{N}"] = {
            "impl": "own" if own_tf > lt_tf else "blaslt",
            "own_tf": round(own_tf, 1), "blaslt_tf": round(lt_tf, 1), "src": name + "_dgrad"}
        nt_total += 1
        own_wins_nt += own_tf > lt_tf
        print(f"{name:10s} dgrad nt M{M} N{K} K{N}: own {own_tf:7.1f} lt {lt_tf:7.1f}")

        # ---- wgrad: tn (x^T @ dy) ------------------------------------------
        _check(C.gemm_bf16_ex(x, dy, 2)[0], x.float().t() @ dy.float(),
               f"{name} wgrad", tol=5e-2)
        t_own = bench(lambda: C.gemm_bf16_ex(x, dy, 2), iters)
        t_lt = bench(lambda: torch.matmul(x.t(), dy), iters)
        own_tf, lt_tf = flops / t_own / 1e12, flops / t_lt / 1e12
        entries[f"tn:{K}x{N}x{M}"] = {
            "impl": "own" if own_tf > lt_tf else "blaslt",
            "own_tf": round(own_tf, 1), "blaslt_tf": round(lt_tf, 1), "src": name + "_wgrad"}
        print(f"{name:10s} wgrad tn K{K} N{N} M{M}: own {own_tf:7.1f} lt {lt_tf:7.1f}")
        del x, w, wt, dy
        torch.cuda.empty_cache()

    # default policy for unmeasured large NT shapes
    default_nt = own_wins_nt >= nt_total * 0.6
    for e in entries.values():
        e["default_nt_own"] = default_nt
    path = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                        "paddle_amd", "ops", "gemm_table.json")
    with open(path, "w") as f:
        json.dump({"version": 1, "device": torch.cuda.get_device_name(0),
                   "entries": entries}, f, indent=1)
    print(f"wrote {path}: {len(entries)} entries, default_nt_own={default_nt}")
    out = os.path.join("gpurun_out", "gemm_table.json")
    os.makedirs("gpurun_out", exist_ok=True)
    with open(out, "w") as f:
        json.dump({"version": 1, "entries": entries}, f, indent=1)


def bench_fp8():
    """fp8 MX kernel vs bf16 hipBLASLt vs torch._scaled_mm at bench shapes."""
    from paddle_amd import _ext
    C = _ext.get_ext()
    shapes = [(16384, 4096, 12288), (16384, 16384, 4096), (8192, 8192, 8192),
              (4096, 4096, 4096), (2048, 14336, 4096)]
    for M, K, N in shapes:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(K, N, device="cuda", dtype=torch.bfloat16) * 0.02
        qx = x.to(torch.float8_e4m3fn)
        qwt = w.t().contiguous().to(torch.float8_e4m3fn)
        flops = 2.0 * M * K * N
        out = C.gemm_fp8_nt(qx, qwt, 1.0)
        ref = qx.float() @ qwt.float().t()
        err = (out.float() - ref).abs().max().item() / ref.abs().max().item()
        t8 = bench(lambda: C.gemm_fp8_nt(qx, qwt, 1.0))
        tb = bench(lambda: torch.matmul(x, w))
        wt_col = w.to(torch.float8_e4m3fn).t().contiguous().t()  # [K,N] col-major
        sc = torch.ones(1, device="cuda")
        tsm = bench(lambda: torch._scaled_mm(qx, wt_col, scale_a=sc, scale_b=sc,
                                             out_dtype=torch.bfloat16))
        print(f"M{M} K{K} N{N}: own-fp8 {flops/t8/1e12:7.1f} TF  "
              f"bf16-lt {flops/tb/1e12:7.1f}  scaled_mm {flops/tsm/1e12:7.1f}  "
              f"(relerr {err:.1e})")
        del x, w, qx, qwt
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
