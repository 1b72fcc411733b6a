"""A/B: experimental 8-phase-class GEMM vs 4-phase own vs hipBLASLt."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from paddle_amd import _ext
C = _ext.get_ext()

def bench(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

for M, N, K in [(4096, 4096, 4096), (8192, 8192, 8192), (24576, 16384, 4096)]:
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.05
    bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.05
    ref = (a[:512].float() @ bt[:512].float().t())
    out8 = C.gemm_bf16_8p(a, bt)
    err = (out8[:512, :512].float() - ref[:, :512]).abs().max() / ref.abs().max()
    fl = 2.0 * M * N * K
    t8 = bench(lambda: C.gemm_bf16_8p(a, bt))
    t4 = bench(lambda: C.gemm_bf16_ex(a, bt, 0)[0])
    tl = bench(lambda: torch.matmul(a, bt.t()))
    print(f"M{M} N{N} K{K}: 8p {fl/t8/1e12:7.1f} TF (relerr {float(err):.3e})  "
          f"4p {fl/t4/1e12:7.1f} TF  lt {fl/tl/1e12:7.1f} TF", flush=True)
    del a, bt, out8
    torch.cuda.empty_cache()
