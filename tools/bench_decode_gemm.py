"""Skinny decode GEMM bench: hipBLASLt vs own split-K vs own MFMA streamer.

L3 is 256 MB, so re-timing ONE weight matrix measures L3-warm reads (the
trap called out in profiles/serving_decode_analysis_r02.txt).  Each op is
timed round-robin over enough weight copies (>1.5 GB) that every read is
HBM-cold, matching the real decode step which cycles 13 GB of weights.
"""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from paddle_amd import _ext
C = _ext.get_ext()

def bench(fn, nw, iters=48):
    for i in range(8): fn(i % nw)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for i in range(iters): fn(i % nw)
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

for m, k, n in [(32, 4096, 12288), (32, 4096, 4096), (32, 4096, 16384),
                (32, 16384, 4096), (16, 4096, 16384), (32, 4096, 50304)]:
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    gb = k * n * 2 / 1e9
    nw = max(2, int(1.6 / gb))          # >1.5 GB of distinct weights
    ws = [torch.randn(k, n, device="cuda", dtype=torch.bfloat16) * 0.02
          for _ in range(nw)]
    t_own = bench(lambda i: C.decode_gemm(x, ws[i], None), nw)
    t_lt = bench(lambda i: torch.matmul(x, ws[i]), nw)
    line = (f"M{m} K{k} N{n}: splitk {t_own*1e6:7.1f}us ({gb/t_own*1e-3:5.2f} TB/s)  "
            f"lt {t_lt*1e6:7.1f}us ({gb/t_lt*1e-3:5.2f} TB/s)")
    if n % 256 == 0 and k % 64 == 0:
        t_mf = bench(lambda i: C.decode_gemm_mfma(x, ws[i], None), nw)
        line += f"  mfma {t_mf*1e6:7.1f}us ({gb/t_mf*1e-3:5.2f} TB/s)"
    print(line, flush=True)
    del ws
    torch.cuda.empty_cache()
