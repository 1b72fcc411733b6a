"""Per-shape GEMM dispatch: own MFMA kernel vs hipBLASLt.

Reference pattern: paddle/phi/kernels/impl/matmul_kernel_impl.h:914-958
(autotune between blas and cublasLt per shape, cached) -- here the table
is measured offline on MI355X (tools/bench_gemm.py --autotune) and
committed as gemm_table.json so dispatch is deterministic at run time.

Layouts (bf16, fp32 accumulate):
  nt:  C[M,N] = A[M,K] @ Bt[N,K]^T   (fwd with cached W^T; dgrad with
       paddle-layout W[K,N] passed as Bt)
  nn:  C[M,N] = A[M,K] @ B[K,N]
  tn:  C[N,K] = At[M,N]^T @ B[M,K]   (wgrad)
"""
from __future__ import annotations

import json
import os

import torch

from .. import _ext

_TABLE = None
_TABLE_PATH = os.path.join(os.path.dirname(__file__), "gemm_table.json")


def _load_table():
    global _TABLE
    if _TABLE is None:
        try:
            with open(_TABLE_PATH) as f:
                _TABLE = json.load(f).get("entries", {})
        except (OSError, ValueError):
            _TABLE = {}
    return _TABLE


def reset_table_cache():
    global _TABLE
    _TABLE = None


def _key(layout, m, n, k):
    return f"{layout}:{m}x{n}x{k}"


def use_own(layout: str, m: int, n: int, k: int) -> bool:
    """True when the hand-written kernel should carry this GEMM."""
    if k % 64 != 0 or m < 512 or n < 512 or k < 512:
        return False
    t = _load_table()
    e = t.get(_key(layout, m, n, k))
    if e is not None:
        return e.get("impl") == "own"
    # unmeasured shape: fall back to the measured-win region.  The 8-phase
    # NT kernel wins on large tile-aligned shapes; stay on hipBLASLt
    # elsewhere until the shape is autotuned.
    if layout == "nt" and m % 256 == 0 and n % 256 == 0 and m >= 4096 and n >= 2048 and k >= 2048:
        for ent in t.values():
            return ent.get("default_nt_own", False) or False
        return False
    return False


def _native_ok(x: torch.Tensor) -> bool:
    return x.is_cuda and x.dtype == torch.bfloat16 and _ext.use_native(x)


# ---------------------------------------------------------------------------
# transposed-weight cache: paddle Linear stores W [in, out]; the fast NT
# kernel wants Bt [out, in].  Cached per (param, version) -- one transpose
# per optimizer step, ~0.5% of the GEMM time it accelerates.
# ---------------------------------------------------------------------------
def weight_t(w: torch.Tensor) -> torch.Tensor:
    cache = getattr(w, "_pa_wt_cache", None)
    ver = w._version
    if cache is not None and cache[0] == ver:
        return cache[1]
    wt = w.t().contiguous()
    try:
        w._pa_wt_cache = (ver, wt)
    except (AttributeError, RuntimeError):
        pass
    return wt


def gemm_nt(a2: torch.Tensor, bt: torch.Tensor, epilogue: int = 0,
            bias=None, aux=None):
    """C = a2 @ bt^T via the own kernel.  Returns tensor (or (C, aux) for
    epilogue==2).  Caller is responsible for the use_own() decision."""
    C = _ext.get_ext()
    outs = C.gemm_bf16_ex(a2, bt, 0, epilogue, bias, aux, None)
    return outs if epilogue == 2 else outs[0]


def gemm_tn(at: torch.Tensor, b: torch.Tensor, c_acc=None):
    """C[N,K] = at^T @ b (wgrad); accumulates into c_acc when given."""
    C = _ext.get_ext()
    return C.gemm_bf16_ex(at, b, 2, 0, None, None, c_acc)[0]
