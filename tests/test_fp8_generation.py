"""fp8 path (CPU fallback math) + paged-cache generation parity."""
import pytest
import torch

import paddle_amd as paddle
from paddle_amd.incubate.fp8 import Fp8Linear, fp8_matmul
from paddle_amd.models import build_gpt
from paddle_amd.models.generation import generate_gpt


def test_fp8_matmul_cpu_fallback_grads():
    x = torch.randn(8, 16).requires_grad_(True)
    w = torch.randn(16, 4).requires_grad_(True)
    out = fp8_matmul(x, w)
    assert out.dtype == torch.bfloat16
    out.float().sum().backward()
    assert x.grad is not None and w.grad is not None


def test_fp8_linear_shapes():
    l = Fp8Linear(32, 8)
    y = l(torch.randn(4, 32, dtype=torch.bfloat16))
    assert y.shape == (4, 8)


def test_generate_matches_full_recompute_cpu():
    """paged-cache greedy decode == full-context forward argmax."""
    paddle.seed(0)
    m = build_gpt("gpt3-tiny")
    m.eval()
    ids = paddle.randint(0, 1024, (2, 12))
    gen = generate_gpt(m, ids, max_new_tokens=5)
    # reference: recompute full forward each step
    cur = ids.clone()
    ref_tokens = []
    with torch.no_grad():
        for _ in range(5):
            logits = m(cur)
            nxt = logits[:, -1].argmax(-1, keepdim=True)
            ref_tokens.append(nxt)
            cur = torch.cat([cur, nxt], dim=1)
    ref = torch.cat(ref_tokens, dim=1)
    assert torch.equal(gen, ref), (gen, ref)


def test_weight_only_quant_roundtrip_cpu():
    import torch
    from paddle_amd import quantization as Q
    torch.manual_seed(5)
    w = torch.randn(128, 64)
    qw, sc = Q.weight_quantize(w)
    x = torch.randn(2, 128)
    out = Q.weight_only_linear(x, qw, sc)
    rel = (out - x @ w).abs().max() / (x @ w).abs().max()
    assert rel < 0.02, float(rel)


def test_convert_linears_to_fp8_trains():
    import torch
    import paddle_amd as paddle
    from paddle_amd.incubate.fp8 import convert_linears_to_fp8
    from paddle_amd.models import GPTPretrainingCriterion, build_gpt
    paddle.seed(0)
    m = build_gpt("gpt3-tiny", max_seq_len=64)
    n = convert_linears_to_fp8(m, min_features=64)
    assert n > 0
    opt = paddle.optimizer.AdamW(learning_rate=1e-3, parameters=m.parameters())
    crit = GPTPretrainingCriterion()
    ids = torch.randint(0, 1024, (2, 32))
    first = last = None
    for _ in range(5):
        loss = crit(m(ids), ids)
        loss.backward()
        opt.step()
        opt.clear_grad()
        first = float(loss) if first is None else first
        last = float(loss)
    assert last < first
