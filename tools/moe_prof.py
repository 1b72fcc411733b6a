"""fp8 vs bf16 grouped-expert FFN microbench (config-5 shapes)."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import paddle_amd as paddle
from paddle_amd.models.moe import GroupedExperts
from paddle_amd.incubate.fp8 import _Fp8GroupedFFN

def bench(fn, iters=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e3

for N in (512, 1024, 2048):
    E, h, I = 64, 1024, 4096
    ge = GroupedExperts(E, h, I)
    for p in ge.parameters():
        p.data = p.data.to("cuda", torch.bfloat16)
    x = torch.randn(E, N, h, device="cuda", dtype=torch.bfloat16)
    xg = x.clone().requires_grad_(True)
    t_bf = bench(lambda: ge(x))
    print("bf16 fwd ok", flush=True)
    t_f8 = bench(lambda: _Fp8GroupedFFN.apply(x, ge.w1, ge.b1, ge.w2, ge.b2))
    print("fp8 fwd ok", flush=True)
    def bf_fb():
        ge(xg).sum().backward()
    def f8_fb():
        _Fp8GroupedFFN.apply(xg, ge.w1, ge.b1, ge.w2, ge.b2).sum().backward()
    t_bfb = bench(bf_fb)
    print("bf16 f+b ok", flush=True)
    t_f8b = bench(f8_fb)
    print("fp8 f+b ok", flush=True)
    print(f"E{E} N{N}: fwd bf16 {t_bf:6.2f}ms fp8 {t_f8:6.2f}ms | "
          f"f+b bf16 {t_bfb:6.2f}ms fp8 {t_f8b:6.2f}ms", flush=True)
    del ge, x, xg
    torch.cuda.empty_cache()
