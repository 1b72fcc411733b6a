import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from paddle_amd import _ext
C = _ext.get_ext()

def bench(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

for m, k, n in [(32, 4096, 12288), (32, 4096, 4096), (32, 4096, 16384),
                (32, 16384, 4096), (16, 4096, 16384), (32, 4096, 50304)]:
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(k, n, device="cuda", dtype=torch.bfloat16) * 0.02
    t_own = bench(lambda: C.decode_gemm(x, w, None))
    t_lt = bench(lambda: torch.matmul(x, w))
    gb = k * n * 2 / 1e9
    print(f"M{m} K{k} N{n}: own {t_own*1e6:7.1f}us ({gb/t_own:6.2f} TB/s)  "
          f"lt {t_lt*1e6:7.1f}us ({gb/t_lt:6.2f} TB/s)")
