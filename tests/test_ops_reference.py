"""Op semantics on the CPU reference path (fp32 torch oracle) --
same functions the HIP kernels are tested against on GPU
(tests/test_kernels_gpu.py).  Gradient checks vs torch autograd."""
import numpy as np
import pytest
import torch

import paddle_amd as paddle
from paddle_amd.ops import functional as hot


def _grad_check(our_fn, ref_fn, *shapes, atol=1e-4):
    xs_our = [torch.randn(s, dtype=torch.float64).float().requires_grad_(True) for s in shapes]
    xs_ref = [x.detach().clone().requires_grad_(True) for x in xs_our]
    out_our = our_fn(*xs_our)
    out_ref = ref_fn(*xs_ref)
    torch.testing.assert_close(out_our, out_ref, rtol=1e-4, atol=atol)
    g = torch.randn_like(out_our)
    out_our.backward(g)
    out_ref.backward(g)
    for a, b in zip(xs_our, xs_ref):
        torch.testing.assert_close(a.grad, b.grad, rtol=1e-4, atol=atol)


def test_layer_norm_fwd_bwd():
    _grad_check(
        lambda x, w, b: hot.layer_norm(x, w, b, 1e-5),
        lambda x, w, b: torch.nn.functional.layer_norm(x, (64,), w, b, 1e-5),
        (4, 64), (64,), (64,))


def test_rms_norm_fwd_bwd():
    def ref(x, w):
        v = x * torch.rsqrt(x.square().mean(-1, keepdim=True) + 1e-6)
        return v * w
    _grad_check(lambda x, w: hot.rms_norm(x, w, 1e-6), ref, (4, 64), (64,))


def test_softmax_cross_entropy():
    logits = torch.randn(8, 100).requires_grad_(True)
    labels = torch.randint(0, 100, (8,))
    loss = hot.softmax_cross_entropy(logits, labels, reduction="mean")
    ref_logits = logits.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(ref_logits, labels)
    torch.testing.assert_close(loss, ref, rtol=1e-5, atol=1e-5)
    loss.backward()
    ref.backward()
    torch.testing.assert_close(logits.grad, ref_logits.grad, rtol=1e-4, atol=1e-5)


def test_softmax_cross_entropy_ignore_index():
    logits = torch.randn(6, 10).requires_grad_(True)
    labels = torch.tensor([1, 2, -100, 4, -100, 0])
    loss = hot.softmax_cross_entropy(logits, labels, ignore_index=-100, reduction="mean")
    ref = torch.nn.functional.cross_entropy(logits.detach(), labels, ignore_index=-100)
    torch.testing.assert_close(loss, ref, rtol=1e-5, atol=1e-5)


def test_bias_gelu():
    _grad_check(
        lambda x, b: hot.bias_gelu(x, b),
        lambda x, b: torch.nn.functional.gelu(x + b),
        (4, 32), (32,))


def test_swiglu():
    def ref(x):
        g, u = x.chunk(2, -1)
        return torch.nn.functional.silu(g) * u
    _grad_check(lambda x: hot.swiglu(x), ref, (4, 64))


def test_rope_matches_reference():
    torch.manual_seed(0)
    b, s, h, d = 2, 16, 4, 32
    q = torch.randn(b, s, h, d).requires_grad_(True)
    out = hot.fused_rotary_position_embedding(q)
    # independent numpy-ish reference
    half = d // 2
    inv = 1.0 / (10000.0 ** (torch.arange(0, half).float() / half))
    t = torch.arange(s).float()
    freqs = torch.outer(t, inv)
    cos = freqs.cos().view(1, s, 1, half)
    sin = freqs.sin().view(1, s, 1, half)
    x1, x2 = q.detach()[..., :half], q.detach()[..., half:]
    ref = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], -1)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    # rotation preserves norms -> grad of sum of squares equals 2q
    (out.square().sum() * 0.5).backward()
    torch.testing.assert_close(q.grad, q.detach(), rtol=1e-4, atol=1e-5)


def test_flash_attention_ref_path_causal():
    torch.manual_seed(0)
    b, s, h, d = 2, 32, 2, 16
    q = torch.randn(b, s, h, d).requires_grad_(True)
    k = torch.randn(b, s, h, d).requires_grad_(True)
    v = torch.randn(b, s, h, d).requires_grad_(True)
    out, _ = hot.flash_attention(q, k, v, causal=True)
    # composed reference
    qt, kt, vt = (t.detach().permute(0, 2, 1, 3) for t in (q, k, v))
    sc = qt @ kt.transpose(-1, -2) / (d ** 0.5)
    mask = torch.ones(s, s, dtype=torch.bool).tril()
    sc = sc.masked_fill(~mask, float("-inf"))
    ref = (torch.softmax(sc, -1) @ vt).permute(0, 2, 1, 3)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)
    out.sum().backward()
    assert q.grad is not None and k.grad is not None and v.grad is not None


def test_dropout_add_zero_p():
    x = torch.randn(4, 8).requires_grad_(True)
    r = torch.randn(4, 8)
    y = hot.dropout_add(x, r, 0.0, True)
    torch.testing.assert_close(y, x + r)


def test_embedding_grad():
    table = torch.randn(10, 8).requires_grad_(True)
    ids = torch.tensor([[1, 2], [2, 3]])
    out = hot.embedding(ids, table)
    out.sum().backward()
    expected = torch.zeros(10, 8)
    for i in [1, 2, 2, 3]:
        expected[i] += 1
    torch.testing.assert_close(table.grad, expected)


def test_fused_adamw_matches_torch():
    torch.manual_seed(0)
    n = 64
    master = torch.randn(n)
    grad = torch.randn(n)
    m = torch.zeros(n)
    v = torch.zeros(n)
    ref = master.clone().requires_grad_(True)
    opt = torch.optim.AdamW([ref], lr=1e-2, betas=(0.9, 0.999), eps=1e-8,
                            weight_decay=0.01)
    ref.grad = grad.clone()
    opt.step()
    hot.fused_adamw_step(master, None, grad, m, v, 1e-2, 0.9, 0.999, 1e-8, 0.01, 1)
    torch.testing.assert_close(master, ref.detach(), rtol=1e-5, atol=1e-6)
