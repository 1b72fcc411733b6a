import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from paddle_amd import quantization as Q

def bench_cycle(fns, iters=30):
    for f in fns[:2]: f()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for i in range(iters): fns[i % len(fns)]()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

for m, k, n in [(16, 4096, 16384), (32, 4096, 16384), (32, 16384, 4096)]:
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    # >L3 working set: 4 weight copies cycled
    ws = [torch.randn(k, n, device="cuda", dtype=torch.bfloat16) * 0.02 for _ in range(4)]
    qs = [Q.weight_quantize(w, algo="weight_only_int8") for w in ws]
    gb = k * n * 2 / 1e9
    t_lt = bench_cycle([(lambda w=w: torch.matmul(x, w)) for w in ws])
    t_q = bench_cycle([(lambda qw=qw, sc=sc: Q.weight_only_linear(x, qw, sc)) for qw, sc in qs])
    # accuracy
    ref = x.float() @ ws[0].float()
    got = Q.weight_only_linear(x, qs[0][0], qs[0][1]).float()
    err = (got - ref).abs().max().item() / ref.abs().max().item()
    print(f"M{m} K{k} N{n}: bf16-lt {t_lt*1e6:7.1f}us ({gb/t_lt*1e-3:5.2f} TB/s)  "
          f"int8-wo {t_q*1e6:7.1f}us ({gb/2/t_q*1e-3:5.2f} TB/s-int8)  relerr {err:.3f}")
