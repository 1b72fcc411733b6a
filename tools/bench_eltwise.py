"""Microbench the HBM-bound elementwise/reduction kernels at bench shapes.

Prints achieved TB/s against the ~8 TB/s HBM3E peak; run on MI355X.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
from paddle_amd.ops import functional as hot  # noqa: E402
import paddle_amd._C as C  # noqa: E402

DEV = "cuda"
torch.manual_seed(0)


def timeit(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def report(name, sec, gb):
    print(f"{name:28s} {sec * 1e6:9.1f} us   {gb / sec / 1e3:6.2f} TB/s")


# GPT-6.7B shapes: tokens = 8*2048, hidden 4096, ffn 16384
M, H, F = 8 * 2048, 4096, 16384
GB = 1e9

z = torch.randn(M, F, device=DEV, dtype=torch.bfloat16)
b = torch.randn(F, device=DEV, dtype=torch.bfloat16)
dy = torch.randn(M, F, device=DEV, dtype=torch.bfloat16)

t = timeit(lambda: C.bias_gelu_fwd(z, b))
report("bias_gelu fwd [16k,16k]", t, 2 * M * F * 2 / GB)

t = timeit(lambda: C.bias_gelu_bwd(dy, z, b))
report("bias_gelu bwd", t, 3 * M * F * 2 / GB)

t = timeit(lambda: C.colsum(dy))
report("colsum [16k,16k]", t, M * F * 2 / GB)

x = torch.randn(M, H, device=DEV, dtype=torch.bfloat16)
w = torch.randn(H, device=DEV, dtype=torch.bfloat16)
bb = torch.randn(H, device=DEV, dtype=torch.bfloat16)
y, mean, rstd = C.layer_norm_fwd(x, w, bb, 1e-5)
dyh = torch.randn(M, H, device=DEV, dtype=torch.bfloat16)

t = timeit(lambda: C.layer_norm_fwd(x, w, bb, 1e-5))
report("ln fwd [16k,4k]", t, 2 * M * H * 2 / GB)

t = timeit(lambda: C.layer_norm_bwd(dyh, x, w, mean, rstd, True))
report("ln bwd (dx+dwdb)", t, 5 * M * H * 2 / GB)

r = torch.randn(M, H, device=DEV, dtype=torch.bfloat16)
t = timeit(lambda: C.dropout_add_fwd(x, r, 0.1, 1234, 0))
report("dropout_add fwd [16k,4k]", t, 3 * M * H * 2 / GB)

g = torch.randn(M, device=DEV)
n = 201 * 2 ** 20
master = torch.randn(n, device=DEV)
grad = torch.randn(n, device=DEV, dtype=torch.bfloat16)
mm = torch.zeros(n, device=DEV)
vv = torch.zeros(n, device=DEV)
pb = torch.empty(n, device=DEV, dtype=torch.bfloat16)
t = timeit(lambda: C.adamw(master, pb, grad, mm, vv, 1e-4, 0.9, 0.95, 1e-8,
                           0.1, 0.9, 0.95, 1.0), iters=10)
report("adamw 201M (bf16 grad)", t, n * (4 * 3 * 2 + 2 + 2) / GB)

t = timeit(lambda: C.l2norm_sq(grad), iters=10)
report("l2norm 201M bf16", t, n * 2 / GB)
