"""GPU numerics: every gfx950 HIP kernel vs a plain torch fp32 reference
(the contract from the task brief: HIP kernel vs fp32 torch oracle).

All marked @pytest.mark.gpu -- run via gpurun / the driver on MI355X.
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    import paddle_amd as paddle
    from paddle_amd import _ext
    from paddle_amd.ops import functional as hot

    DEV = "cuda:0"


def _bf(x):
    return x.to(torch.bfloat16)


def _assert_close_bf16(ours, ref_fp32, atol=2e-2, rtol=2e-2):
    torch.testing.assert_close(ours.float(), ref_fp32.float(), atol=atol, rtol=rtol)


def test_extension_loaded():
    assert _ext.has_ext(), "gfx950 extension must be built+loaded on GPU"
    C = _ext.get_ext()
    assert C.compiled_arch == "gfx950"


def test_mfma_fragment_layout_probe():
    """Asymmetric-input MFMA probe (guide G9): validates the A/B/C lane
    mappings used by flash_attn.hip."""
    C = _ext.get_ext()
    torch.manual_seed(0)
    a = torch.randn(16, 32, device=DEV)
    b = torch.randn(32, 16, device=DEV)
    out = C.mfma_probe(_bf(a).contiguous(), _bf(b.t()).contiguous())
    ref = _bf(a).float() @ _bf(b).float()
    torch.testing.assert_close(out, ref, atol=1e-2, rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_layer_norm_gpu(dtype):
    torch.manual_seed(0)
    x = torch.randn(128, 4096, device=DEV, dtype=dtype).requires_grad_(True)
    w = torch.randn(4096, device=DEV, dtype=dtype).requires_grad_(True)
    b = torch.randn(4096, device=DEV, dtype=dtype).requires_grad_(True)
    y = hot.layer_norm(x, w, b, 1e-5)
    ref = torch.nn.functional.layer_norm(x.detach().float(), (4096,),
                                         w.detach().float(), b.detach().float(), 1e-5)
    _assert_close_bf16(y, ref)
    g = torch.randn_like(y)
    y.backward(g)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True)
    torch.nn.functional.layer_norm(xr, (4096,), wr, br, 1e-5).backward(g.float())
    _assert_close_bf16(x.grad, xr.grad, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(w.grad, wr.grad, atol=0.2, rtol=2e-2)
    _assert_close_bf16(b.grad, br.grad, atol=0.2, rtol=2e-2)


def test_rms_norm_gpu():
    torch.manual_seed(1)
    x = torch.randn(256, 2048, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    w = torch.randn(2048, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    y = hot.rms_norm(x, w, 1e-6)
    xf = x.detach().float()
    ref = xf * torch.rsqrt(xf.square().mean(-1, keepdim=True) + 1e-6) * w.detach().float()
    _assert_close_bf16(y, ref)
    g = torch.randn_like(y)
    y.backward(g)
    xr = xf.requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    (xr * torch.rsqrt(xr.square().mean(-1, keepdim=True) + 1e-6) * wr).backward(g.float())
    _assert_close_bf16(x.grad, xr.grad, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(w.grad, wr.grad, atol=0.5, rtol=2e-2)


def test_fused_rms_norm_residual_gpu():
    x = torch.randn(64, 1024, device=DEV, dtype=torch.bfloat16)
    res = torch.randn_like(x)
    w = torch.ones(1024, device=DEV, dtype=torch.bfloat16)
    y, res_out = hot.fused_rms_norm(x, w, residual=res)
    xr = (x + res).float()
    ref = xr * torch.rsqrt(xr.square().mean(-1, keepdim=True) + 1e-6)
    _assert_close_bf16(res_out, xr)
    _assert_close_bf16(y, ref)


def test_softmax_cross_entropy_gpu():
    torch.manual_seed(2)
    n, v = 512, 50304
    logits = torch.randn(n, v, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    labels = torch.randint(0, v, (n,), device=DEV)
    labels[::7] = -100
    loss = hot.softmax_cross_entropy(logits, labels, ignore_index=-100, reduction="mean")
    lr = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lr, labels, ignore_index=-100)
    torch.testing.assert_close(loss.float(), ref, atol=2e-2, rtol=2e-2)
    loss.backward()
    ref.backward()
    _assert_close_bf16(logits.grad, lr.grad, atol=1e-3, rtol=5e-2)


@pytest.mark.parametrize("causal", [True, False])
@pytest.mark.parametrize("shape", [(2, 4, 256, 128), (1, 2, 512, 64), (2, 3, 200, 128)])
def test_flash_attention_gpu(causal, shape):
    torch.manual_seed(3)
    b, h, s, d = shape
    q = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    k = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    v = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    from paddle_amd.ops.functional import _FlashAttn, _sdpa_ref, _sdpa_ref_bwd
    o, _lse = _FlashAttn.apply(q, k, v, scale, causal)
    ref_o, ref_lse = _sdpa_ref(q.detach().float(), k.detach().float(),
                               v.detach().float(), scale, causal)
    _assert_close_bf16(o, ref_o, atol=3e-2, rtol=3e-2)
    g = torch.randn_like(o)
    o.backward(g)
    dq, dk, dv = _sdpa_ref_bwd(g.float(), q.detach().float(), k.detach().float(),
                               v.detach().float(), ref_lse, scale, causal)
    _assert_close_bf16(q.grad, dq, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(k.grad, dk, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(v.grad, dv, atol=5e-2, rtol=5e-2)


def test_flash_attention_gqa_fwd():
    torch.manual_seed(4)
    b, hq, hkv, s, d = 2, 8, 2, 256, 128
    q = torch.randn(b, hq, s, d, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(b, hkv, s, d, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device=DEV, dtype=torch.bfloat16)
    from paddle_amd.ops.functional import _FlashAttn, _sdpa_ref
    with torch.no_grad():
        o, _lse = _FlashAttn.apply(q, k, v, 1.0 / math.sqrt(d), True)
    ref_o, _ = _sdpa_ref(q.float(), k.float(), v.float(), 1.0 / math.sqrt(d), True)
    _assert_close_bf16(o, ref_o, atol=3e-2, rtol=3e-2)


def test_bias_gelu_swiglu_gpu():
    x = torch.randn(64, 4096, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    b = torch.randn(4096, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    y = hot.bias_gelu(x, b)
    ref = torch.nn.functional.gelu(x.detach().float() + b.detach().float())
    _assert_close_bf16(y, ref)
    y.sum().backward()
    assert x.grad is not None and b.grad is not None

    x2 = torch.randn(64, 2048, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    y2 = hot.swiglu(x2)
    g, u = x2.detach().float().chunk(2, -1)
    _assert_close_bf16(y2, torch.nn.functional.silu(g) * u)
    y2.sum().backward()
    assert x2.grad is not None


def test_rope_gpu():
    b, s, h, d = 2, 128, 8, 128
    q = torch.randn(b, s, h, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    out = hot.fused_rotary_position_embedding(q)
    from paddle_amd.ops.functional import _rope_ref, build_rope_cache
    cos_t, sin_t = build_rope_cache(s, d, 10000.0, torch.device(DEV))
    ref = _rope_ref(q.detach().float(), cos_t, sin_t, 0, False)
    _assert_close_bf16(out, ref)
    (out.float().square().sum() * 0.5).backward()
    # rotation preserves norm: d/dq 0.5*|R q|^2 = q
    _assert_close_bf16(q.grad, q.detach(), atol=3e-2, rtol=3e-2)


def test_embedding_gpu():
    table = torch.randn(50304, 1024, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    ids = torch.randint(0, 50304, (4, 512), device=DEV)
    out = hot.embedding(ids, table)
    ref = torch.nn.functional.embedding(ids, table.detach().float())
    _assert_close_bf16(out, ref)
    g = torch.randn_like(out)
    out.backward(g)
    tr = table.detach().float().requires_grad_(True)
    torch.nn.functional.embedding(ids, tr).backward(g.float())
    _assert_close_bf16(table.grad, tr.grad, atol=5e-2, rtol=5e-2)


def test_dropout_add_gpu():
    x = torch.randn(1024, 1024, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    r = torch.randn_like(x)
    y = hot.dropout_add(x, r, 0.1, True)
    # statistical: mean of (y - r) ~ mean of x
    kept = ((y - r).abs() > 1e-6).float().mean()
    assert 0.85 < float(kept) < 0.95
    y.sum().backward()
    assert x.grad is not None
    # p=0 exact
    y0 = hot.dropout_add(x, r, 0.0, True)
    _assert_close_bf16(y0, (x + r).float())


def test_fused_adamw_gpu():
    torch.manual_seed(5)
    n = 4096
    master = torch.randn(n, device=DEV)
    grad = torch.randn(n, device=DEV, dtype=torch.bfloat16)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    param_bf16 = master.to(torch.bfloat16)
    ref = master.clone().requires_grad_(True)
    opt = torch.optim.AdamW([ref], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.1)
    ref.grad = grad.float()
    opt.step()
    hot.fused_adamw_step(master, param_bf16, grad, m, v, 1e-2, 0.9, 0.95, 1e-8, 0.1, 1)
    torch.testing.assert_close(master, ref.detach(), atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(param_bf16.float(), master, atol=1e-2, rtol=1e-2)


def test_fused_adamw_tail_gpu():
    # numel not divisible by 8 (per-param path, e.g. a 2-elem bias)
    for n in (2, 13, 4099):
        master = torch.randn(n, device=DEV)
        grad = torch.randn(n, device=DEV, dtype=torch.bfloat16)
        m = torch.zeros(n, device=DEV)
        v = torch.zeros(n, device=DEV)
        ref = master.clone().requires_grad_(True)
        opt = torch.optim.AdamW([ref], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                                weight_decay=0.1)
        ref.grad = grad.float()
        opt.step()
        hot.fused_adamw_step(master, None, grad, m, v, 1e-2, 0.9, 0.95, 1e-8, 0.1, 1)
        torch.testing.assert_close(master, ref.detach(), atol=1e-5, rtol=1e-5)


def test_fused_adamw_grad_scale_gpu():
    # grad_scale folds clip/1-world into the kernel: scaled bf16 grads must
    # match pre-scaled fp32 grads through torch.optim.AdamW
    torch.manual_seed(6)
    n = 8192
    master = torch.randn(n, device=DEV)
    grad = torch.randn(n, device=DEV, dtype=torch.bfloat16)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    ref = master.clone().requires_grad_(True)
    opt = torch.optim.AdamW([ref], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.1)
    ref.grad = grad.float() * 0.37
    opt.step()
    hot.fused_adamw_step(master, None, grad, m, v, 1e-2, 0.9, 0.95, 1e-8,
                         0.1, 1, grad_scale=0.37)
    torch.testing.assert_close(master, ref.detach(), atol=1e-5, rtol=1e-5)


def test_l2norm_gpu():
    x = torch.randn(123456, device=DEV, dtype=torch.bfloat16)
    out = hot.l2_norm_squared(x)
    ref = x.float().square().sum()
    torch.testing.assert_close(out[0], ref, rtol=1e-2, atol=1.0)


def test_colsum_gpu():
    C = _ext.get_ext()
    x = torch.randn(512, 1024, device=DEV, dtype=torch.bfloat16)
    out = C.colsum(x)
    torch.testing.assert_close(out, x.float().sum(0), rtol=1e-2, atol=0.5)


def test_paged_decode_attention_gpu():
    torch.manual_seed(7)
    B, H, HKV, D, bs, max_blocks = 4, 8, 2, 128, 16, 16
    nblocks = 64
    q = torch.randn(B, H, D, device=DEV, dtype=torch.bfloat16)
    kc = torch.randn(nblocks, bs, HKV, D, device=DEV, dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    bt = torch.randperm(nblocks, device=DEV)[: B * max_blocks].reshape(B, max_blocks).int()
    seq = torch.tensor([7, 16, 100, 250], device=DEV, dtype=torch.int32)
    out = hot.paged_decode_attention(q, kc, vc, bt, seq)
    # reference path (force fallback)
    import paddle_amd
    paddle_amd.set_flags({"FLAGS_use_native_kernels": False})
    ref = hot.paged_decode_attention(q, kc, vc, bt, seq)
    paddle_amd.set_flags({"FLAGS_use_native_kernels": True})
    _assert_close_bf16(out, ref.float(), atol=3e-2, rtol=3e-2)


def test_fused_ffn_hipblaslt_gpu():
    """fc1 bias+GELU epilogue + fc2, backward with DGELU_BGRAD epilogue,
    vs fp32 torch reference (tanh-approx GELU)."""
    if not hot._fused_ffn_available(torch.zeros(1, device=DEV, dtype=torch.bfloat16)):
        pytest.skip("hipBLASLt build lacks GELU aux epilogues (probe)")
    torch.manual_seed(11)
    M, K, N, H = 512, 256, 1024, 256
    x = torch.randn(M, K, device=DEV, dtype=torch.bfloat16, requires_grad=True)
    w1 = torch.randn(K, N, device=DEV, dtype=torch.bfloat16) * 0.05
    b1 = torch.randn(N, device=DEV, dtype=torch.bfloat16) * 0.1
    w2 = torch.randn(N, H, device=DEV, dtype=torch.bfloat16) * 0.05
    b2 = torch.randn(H, device=DEV, dtype=torch.bfloat16) * 0.1
    for t in (w1, b1, w2, b2):
        t.requires_grad_(True)
    y = hot.fused_ffn(x, w1, b1, w2, b2)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    w1f = w1.detach().float().requires_grad_(True)
    b1f = b1.detach().float().requires_grad_(True)
    w2f = w2.detach().float().requires_grad_(True)
    b2f = b2.detach().float().requires_grad_(True)
    g = torch.nn.functional.gelu(xf @ w1f + b1f, approximate="tanh")
    yr = g @ w2f + b2f
    yr.backward(dy.float())

    torch.testing.assert_close(y.float(), yr, atol=0.12, rtol=0.05)
    torch.testing.assert_close(x.grad.float(), xf.grad, atol=0.15, rtol=0.08)
    torch.testing.assert_close(b1.grad.float(), b1f.grad, atol=0.8, rtol=0.05)
    torch.testing.assert_close(b2.grad.float(), b2f.grad, atol=0.8, rtol=0.05)
    torch.testing.assert_close(w1.grad.float(), w1f.grad, atol=0.8, rtol=0.08)
    torch.testing.assert_close(w2.grad.float(), w2f.grad, atol=0.8, rtol=0.08)


@pytest.mark.parametrize("m,k,n", [(4, 4096, 1024), (32, 4160, 4096),
                                   (16, 512, 256)])
def test_weight_only_decode_gpu(m, k, n):
    """int8 weight-only decode (MFMA W-streamer) vs dequantized matmul."""
    from paddle_amd import quantization as Q
    torch.manual_seed(13)
    w = torch.randn(k, n, device=DEV) * 0.1
    b = torch.randn(n, device=DEV)
    qw, sc = Q.weight_quantize(w)
    x = torch.randn(m, k, device=DEV, dtype=torch.bfloat16)
    out = Q.weight_only_linear(x, qw.to(DEV), sc.to(DEV),
                               bias=b.to(torch.bfloat16))
    wref = qw.to(DEV).float() * sc.to(DEV).unsqueeze(0) / 127.0
    ref = x.float() @ wref + b
    torch.testing.assert_close(out.float(), ref, atol=0.5, rtol=0.05)


@pytest.mark.parametrize("sq,skv", [(128, 384), (384, 128), (100, 260)])
def test_flash_attention_cross_gpu(sq, skv):
    """Cross-attention (Sq != Skv, non-causal) fwd+bwd vs fp32 reference."""
    torch.manual_seed(4)
    b, h, d = 2, 4, 128
    q = torch.randn(b, h, sq, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    k = torch.randn(b, h, skv, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    v = torch.randn(b, h, skv, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    from paddle_amd.ops.functional import _FlashAttn, _sdpa_ref, _sdpa_ref_bwd
    o, _lse = _FlashAttn.apply(q, k, v, scale, False)
    ref_o, ref_lse = _sdpa_ref(q.detach().float(), k.detach().float(),
                               v.detach().float(), scale, False)
    _assert_close_bf16(o, ref_o, atol=3e-2, rtol=3e-2)
    g = torch.randn_like(o)
    o.backward(g)
    dq, dk, dv = _sdpa_ref_bwd(g.float(), q.detach().float(), k.detach().float(),
                               v.detach().float(), ref_lse, scale, False)
    _assert_close_bf16(q.grad, dq, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(k.grad, dk, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(v.grad, dv, atol=5e-2, rtol=5e-2)


def test_flash_attn_varlen_gpu():
    """Ragged flash attention vs per-sequence reference."""
    from paddle_amd.ops.functional import flash_attn_varlen_func, _sdpa_ref
    torch.manual_seed(9)
    H, D = 4, 128
    lens = [96, 160, 96]
    cu = torch.tensor([0, 96, 256, 352], dtype=torch.int32, device=DEV)
    total = sum(lens)
    q = torch.randn(total, H, D, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(total, H, D, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(total, H, D, device=DEV, dtype=torch.bfloat16)
    out = flash_attn_varlen_func(q, k, v, cu, cu, max(lens), max(lens),
                                 causal=True)
    off = 0
    for L in lens:
        qs = q[off:off + L].transpose(0, 1).unsqueeze(0).float()
        ks = k[off:off + L].transpose(0, 1).unsqueeze(0).float()
        vs = v[off:off + L].transpose(0, 1).unsqueeze(0).float()
        ref, _ = _sdpa_ref(qs, ks, vs, 1.0 / math.sqrt(D), True)
        got = out[off:off + L].transpose(0, 1).unsqueeze(0).float()
        torch.testing.assert_close(got, ref, atol=3e-2, rtol=3e-2)
        off += L


def test_flash_attention_gqa_bwd_gpu():
    """GQA (H > HKV) backward: expanded-KV kernel + head-group reduction of
    dK/dV vs fp32 reference."""
    torch.manual_seed(21)
    b, hq, hkv, s, d = 2, 8, 2, 192, 128
    q = torch.randn(b, hq, s, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    k = torch.randn(b, hkv, s, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    v = torch.randn(b, hkv, s, d, device=DEV, dtype=torch.bfloat16).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    from paddle_amd.ops.functional import _FlashAttn
    o, _lse = _FlashAttn.apply(q, k, v, scale, True)
    g = torch.randn_like(o)
    o.backward(g)
    rep = hq // hkv
    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    ke = kf.repeat_interleave(rep, dim=1)
    ve = vf.repeat_interleave(rep, dim=1)
    sc = (qf @ ke.transpose(-1, -2)) * scale
    mask = torch.triu(torch.ones(s, s, device=DEV, dtype=torch.bool), 1)
    sc = sc.masked_fill(mask, float("-inf"))
    ref = torch.softmax(sc, -1) @ ve
    ref.backward(g.float())
    _assert_close_bf16(o, ref, atol=3e-2, rtol=3e-2)
    _assert_close_bf16(q.grad, qf.grad, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(k.grad, kf.grad, atol=8e-2, rtol=8e-2)
    _assert_close_bf16(v.grad, vf.grad, atol=8e-2, rtol=8e-2)


# ---------------------------------------------------------------------------
# gemm.hip: 8-phase MFMA GEMM, all layouts + epilogues (vs fp32 torch)
# ---------------------------------------------------------------------------
def _gemm_rel_ok(out, ref, tol=3e-2):
    err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
    assert err < tol, f"relerr {err:.3e}"


@pytest.mark.parametrize("m,n,k", [(512, 512, 512), (768, 1024, 256),
                                   (2048, 2048, 4096)])
def test_gemm_nt_numerics(m, n, k):
    torch.manual_seed(1)
    C = _ext.get_ext()
    a = _bf(torch.randn(m, k, device=DEV))
    bt = _bf(torch.randn(n, k, device=DEV))
    out = C.gemm_bf16_ex(a, bt, 0)[0]
    _gemm_rel_ok(out, a.float() @ bt.float().t())


def test_amax_abs():
    C = _ext.get_ext()
    for n in (999, 4096, 1 << 20):
        x = _bf(torch.randn(n, device=DEV) * 3)
        ref = x.float().abs().amax()
        out = C.amax_abs(x)
        assert torch.allclose(out, ref), (float(out), float(ref))


def test_gemm_fp8_batched_bias():
    torch.manual_seed(7)
    C = _ext.get_ext()
    e, m, n, k = 8, 256, 512, 256
    a = torch.randn(e, m, k, device=DEV).clamp(-2, 2)
    bt = torch.randn(e, n, k, device=DEV).clamp(-2, 2)
    bias = _bf(torch.randn(e, n, device=DEV))
    qa = (a * 16).to(torch.float8_e4m3fn)
    qb = (bt * 16).to(torch.float8_e4m3fn)
    s = 1.0 / 256
    out = C.gemm_fp8_nt_batched(qa, qb, s, bias)
    ref = torch.einsum("emk,enk->emn", qa.float(), qb.float()) * s \
        + bias.float().unsqueeze(1)
    _gemm_rel_ok(out, ref, tol=5e-2)


@pytest.mark.parametrize("e,m,n,k", [(4, 512, 512, 512), (8, 513, 1024, 4096),
                                     (64, 513, 4096, 1024)])
def test_gemm_nt_batched(e, m, n, k):
    torch.manual_seed(2)
    C = _ext.get_ext()
    a = _bf(torch.randn(e, m, k, device=DEV))
    bt = _bf(torch.randn(e, n, k, device=DEV))
    out = C.gemm_bf16_nt_batched(a, bt)
    ref = torch.einsum("emk,enk->emn", a.float(), bt.float())
    _gemm_rel_ok(out, ref)


@pytest.mark.parametrize("m,n,k", [(512, 512, 512), (2048, 1024, 2048)])
def test_gemm_nn_numerics(m, n, k):
    torch.manual_seed(2)
    C = _ext.get_ext()
    a = _bf(torch.randn(m, k, device=DEV))
    b = _bf(torch.randn(k, n, device=DEV))
    out = C.gemm_bf16_ex(a, b, 1)[0]
    _gemm_rel_ok(out, a.float() @ b.float())


@pytest.mark.parametrize("m,n,k", [(512, 512, 512), (1024, 2048, 2048)])
def test_gemm_tn_numerics(m, n, k):
    # wgrad: C[m,n] = At[k,m]^T @ B[k,n], reduction over k (the token dim)
    torch.manual_seed(3)
    C = _ext.get_ext()
    at = _bf(torch.randn(k, m, device=DEV))
    b = _bf(torch.randn(k, n, device=DEV))
    out = C.gemm_bf16_ex(at, b, 2)[0]
    _gemm_rel_ok(out, at.float().t() @ b.float(), tol=5e-2)


def test_gemm_boundary_tiles():
    # M/N not multiples of 256 exercise the guarded kernel + skip-interior
    torch.manual_seed(4)
    C = _ext.get_ext()
    for (m, n, k) in [(300, 520, 512), (512, 777, 256), (130, 200, 128),
                      (1000, 50304 % 2048 + 304, 512)]:
        a = _bf(torch.randn(m, k, device=DEV))
        bt = _bf(torch.randn(n, k, device=DEV))
        out = C.gemm_bf16_ex(a, bt, 0)[0]
        _gemm_rel_ok(out, a.float() @ bt.float().t())


def test_gemm_k_tail():
    # K not a multiple of 64: whole grid takes the guarded path
    torch.manual_seed(5)
    C = _ext.get_ext()
    a = _bf(torch.randn(512, 200, device=DEV))
    bt = _bf(torch.randn(512, 200, device=DEV))
    out = C.gemm_bf16_ex(a, bt, 0)[0]
    _gemm_rel_ok(out, a.float() @ bt.float().t())


def test_gemm_bias_epilogue():
    torch.manual_seed(6)
    C = _ext.get_ext()
    a = _bf(torch.randn(512, 512, device=DEV))
    bt = _bf(torch.randn(768, 512, device=DEV))
    bias = _bf(torch.randn(768, device=DEV))
    out = C.gemm_bf16_ex(a, bt, 0, 1, bias)[0]
    _gemm_rel_ok(out, a.float() @ bt.float().t() + bias.float())


def test_gemm_bias_gelu_epilogue():
    torch.manual_seed(7)
    C = _ext.get_ext()
    a = _bf(torch.randn(512, 512, device=DEV) * 0.5)
    bt = _bf(torch.randn(768, 512, device=DEV) * 0.05)
    bias = _bf(torch.randn(768, device=DEV) * 0.1)
    out, aux = C.gemm_bf16_ex(a, bt, 0, 2, bias)
    pre = a.float() @ bt.float().t() + bias.float()
    _gemm_rel_ok(aux, pre)
    _gemm_rel_ok(out, torch.nn.functional.gelu(pre), tol=4e-2)


def test_gemm_dgelu_epilogue():
    torch.manual_seed(8)
    C = _ext.get_ext()
    dy = _bf(torch.randn(512, 512, device=DEV))
    w = _bf(torch.randn(768, 512, device=DEV) * 0.05)   # NT B-operand [n,k]
    z = _bf(torch.randn(512, 768, device=DEV))          # saved pre-act
    out = C.gemm_bf16_ex(dy, w, 0, 3, None, z)[0]       # [512, 768]
    zf = z.float().requires_grad_(True)
    torch.nn.functional.gelu(zf).backward(dy.float() @ w.float().t())
    # out = (dy @ w^T) * gelu'(z)
    _gemm_rel_ok(out, zf.grad, tol=5e-2)


def test_gemm_accumulate():
    torch.manual_seed(9)
    C = _ext.get_ext()
    at = _bf(torch.randn(512, 512, device=DEV))
    b = _bf(torch.randn(512, 768, device=DEV))
    c0 = _bf(torch.randn(512, 768, device=DEV))
    acc = c0.clone()
    C.gemm_bf16_ex(at, b, 2, 0, None, None, acc)
    _gemm_rel_ok(acc, c0.float() + at.float().t() @ b.float(), tol=5e-2)


def test_fused_linear_own_grads():
    """fused_linear_own fwd+bwd vs fp32 autograd reference."""
    from paddle_amd.ops import gemm_dispatch
    torch.manual_seed(10)
    m, k, n = 1024, 512, 768
    x = _bf(torch.randn(2, m // 2, k, device=DEV)).requires_grad_(True)
    w = (_bf(torch.randn(k, n, device=DEV)) * 0.05).requires_grad_(True)
    bias = (_bf(torch.randn(n, device=DEV)) * 0.1).requires_grad_(True)
    y = hot.fused_linear_own(x, w, bias)
    loss = (y.float() ** 2).mean()
    loss.backward()
    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf_ = bias.detach().float().requires_grad_(True)
    yf = xf @ wf + bf_
    ((yf ** 2).mean()).backward()
    _gemm_rel_ok(y, yf.detach(), tol=4e-2)
    _gemm_rel_ok(x.grad, xf.grad, tol=5e-2)
    _gemm_rel_ok(w.grad, wf.grad, tol=5e-2)
    _gemm_rel_ok(bias.grad, bf_.grad, tol=5e-2)


def test_fused_ffn_own_grads():
    torch.manual_seed(11)
    m, h, ffn = 512, 512, 1024
    x = _bf(torch.randn(m, h, device=DEV)).requires_grad_(True)
    w1 = (_bf(torch.randn(h, ffn, device=DEV)) * 0.05).requires_grad_(True)
    b1 = (_bf(torch.randn(ffn, device=DEV)) * 0.1).requires_grad_(True)
    w2 = (_bf(torch.randn(ffn, h, device=DEV)) * 0.05).requires_grad_(True)
    b2 = (_bf(torch.randn(h, device=DEV)) * 0.1).requires_grad_(True)
    y = hot.fused_ffn_own(x, w1, b1, w2, b2)
    (y.float() ** 2).mean().backward()
    xf = x.detach().float().requires_grad_(True)
    w1f = w1.detach().float().requires_grad_(True)
    b1f = b1.detach().float().requires_grad_(True)
    w2f = w2.detach().float().requires_grad_(True)
    b2f = b2.detach().float().requires_grad_(True)
    yf = torch.nn.functional.gelu(xf @ w1f + b1f) @ w2f + b2f
    (yf ** 2).mean().backward()
    _gemm_rel_ok(y, yf.detach(), tol=4e-2)
    _gemm_rel_ok(x.grad, xf.grad, tol=6e-2)
    _gemm_rel_ok(w1.grad, w1f.grad, tol=6e-2)
    _gemm_rel_ok(b1.grad, b1f.grad, tol=6e-2)
    _gemm_rel_ok(w2.grad, w2f.grad, tol=6e-2)
    _gemm_rel_ok(b2.grad, b2f.grad, tol=6e-2)


def test_weight_t_cache_invalidation():
    from paddle_amd.ops import gemm_dispatch as gd
    w = _bf(torch.randn(64, 32, device=DEV))
    t1 = gd.weight_t(w)
    assert gd.weight_t(w) is t1          # cached
    w.add_(1.0)                          # version bump
    t2 = gd.weight_t(w)
    assert t2 is not t1
    torch.testing.assert_close(t2, w.t().contiguous())


# ---------------------------------------------------------------------------
# flash attention: mask + Philox dropout
# ---------------------------------------------------------------------------
def test_flash_attention_mask_gpu():
    torch.manual_seed(11)
    b, h, s, d = 2, 4, 256, 128
    q = _bf(torch.randn(b, h, s, d, device=DEV)).requires_grad_(True)
    k = _bf(torch.randn(b, h, s, d, device=DEV)).requires_grad_(True)
    v = _bf(torch.randn(b, h, s, d, device=DEV)).requires_grad_(True)
    # additive mask with random -inf blocks + finite biases
    m = torch.zeros(b, 1, s, s, device=DEV)
    m[:, :, :, ::7] = -10000.0
    m[:, :, ::5, :] += 0.25
    m = _bf(m)
    scale = 1.0 / math.sqrt(d)
    from paddle_amd.ops.functional import _FlashAttn, _sdpa_ref, _sdpa_ref_bwd
    o, lse = _FlashAttn.apply(q, k, v, scale, False, m, 0.0, 0, 0)
    ref_o, ref_lse = _sdpa_ref(q.detach().float(), k.detach().float(),
                               v.detach().float(), scale, False, m.float())
    _assert_close_bf16(o, ref_o, atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(lse, ref_lse, atol=2e-2, rtol=2e-2)
    g = torch.randn_like(o)
    o.backward(g)
    dq, dk, dv = _sdpa_ref_bwd(g.float(), q.detach().float(), k.detach().float(),
                               v.detach().float(), ref_lse, scale, False, m.float())
    _assert_close_bf16(q.grad, dq, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(k.grad, dk, atol=5e-2, rtol=5e-2)
    _assert_close_bf16(v.grad, dv, atol=5e-2, rtol=5e-2)


def test_flash_attention_dropout_gpu():
    """Exact check: extract the kernel's Philox keep-mask and feed it to the
    fp32 oracle -- fwd and all three grads must match; plus determinism."""
    torch.manual_seed(12)
    C = _ext.get_ext()
    b, h, s, d = 1, 2, 256, 128
    p, seed, offset = 0.3, 1234, 999
    q = _bf(torch.randn(b, h, s, d, device=DEV)).requires_grad_(True)
    k = _bf(torch.randn(b, h, s, d, device=DEV)).requires_grad_(True)
    v = _bf(torch.randn(b, h, s, d, device=DEV)).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    from paddle_amd.ops.functional import _FlashAttn, _sdpa_ref, _sdpa_ref_bwd
    o, lse = _FlashAttn.apply(q, k, v, scale, False, None, p, seed, offset)
    o2, _ = _FlashAttn.apply(q, k, v, scale, False, None, p, seed, offset)
    torch.testing.assert_close(o, o2)           # recompute-deterministic
    keep = C.fa_dropout_mask(b, h, s, s, p, seed, offset)
    frac = keep.float().mean().item()
    assert abs(frac - (1 - p)) < 0.02, frac     # keep rate sane
    ref_o, ref_lse = _sdpa_ref(q.detach().float(), k.detach().float(),
                               v.detach().float(), scale, False, None,
                               keep, p)
    _assert_close_bf16(o, ref_o, atol=4e-2, rtol=4e-2)
    g = torch.randn_like(o)
    o.backward(g)
    dq, dk, dv = _sdpa_ref_bwd(g.float(), q.detach().float(), k.detach().float(),
                               v.detach().float(), ref_lse, scale, False, None,
                               keep, p)
    _assert_close_bf16(q.grad, dq, atol=6e-2, rtol=6e-2)
    _assert_close_bf16(k.grad, dk, atol=6e-2, rtol=6e-2)
    _assert_close_bf16(v.grad, dv, atol=6e-2, rtol=6e-2)


def test_flash_attention_causal_mask_combo_gpu():
    torch.manual_seed(13)
    b, h, s, d = 1, 2, 200, 64
    q = _bf(torch.randn(b, h, s, d, device=DEV))
    k = _bf(torch.randn(b, h, s, d, device=DEV))
    v = _bf(torch.randn(b, h, s, d, device=DEV))
    m = _bf(torch.randn(b, h, s, s, device=DEV) * 0.5)
    scale = 1.0 / math.sqrt(d)
    from paddle_amd.ops.functional import _FlashAttn, _sdpa_ref
    with torch.no_grad():
        o, _ = _FlashAttn.apply(q, k, v, scale, True, m, 0.0, 0, 0)
    ref_o, _ = _sdpa_ref(q.float(), k.float(), v.float(), scale, True, m.float())
    _assert_close_bf16(o, ref_o, atol=3e-2, rtol=3e-2)


def test_flash_attn_varlen_single_launch_gpu():
    """Mixed lengths, ONE kernel launch: fwd+bwd vs the
    per-sequence fp32 oracle."""
    torch.manual_seed(21)
    from paddle_amd.ops.functional import flash_attn_varlen_func, _sdpa_ref, _sdpa_ref_bwd
    h, d = 4, 128
    lens = [128, 37, 256, 200]
    cu = torch.tensor([0] + list(torch.cumsum(torch.tensor(lens), 0)),
                      dtype=torch.int32, device=DEV)
    total = sum(lens)
    q = _bf(torch.randn(total, h, d, device=DEV)).requires_grad_(True)
    k = _bf(torch.randn(total, h, d, device=DEV)).requires_grad_(True)
    v = _bf(torch.randn(total, h, d, device=DEV)).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    for causal in (False, True):
        if q.grad is not None:
            q.grad = None; k.grad = None; v.grad = None
        out = flash_attn_varlen_func(q, k, v, cu, cu, max(lens), max(lens),
                                     scale=scale, causal=causal)
        g = torch.randn_like(out)
        out.backward(g)
        # per-sequence reference
        o_ref = torch.empty(total, h, d, device=DEV)
        dq_ref = torch.zeros(total, h, d, device=DEV)
        dk_ref = torch.zeros_like(dq_ref)
        dv_ref = torch.zeros_like(dq_ref)
        for i, L in enumerate(lens):
            s0, s1 = int(cu[i]), int(cu[i + 1])
            qs = q.detach()[s0:s1].transpose(0, 1).unsqueeze(0)  # [1,h,L,d]
            ks = k.detach()[s0:s1].transpose(0, 1).unsqueeze(0)
            vs = v.detach()[s0:s1].transpose(0, 1).unsqueeze(0)
            oo, lse = _sdpa_ref(qs.float(), ks.float(), vs.float(), scale, causal)
            o_ref[s0:s1] = oo.squeeze(0).transpose(0, 1)
            gs = g[s0:s1].transpose(0, 1).unsqueeze(0).float()
            dqs, dks, dvs = _sdpa_ref_bwd(gs, qs.float(), ks.float(), vs.float(),
                                          lse, scale, causal)
            dq_ref[s0:s1] = dqs.squeeze(0).transpose(0, 1)
            dk_ref[s0:s1] = dks.squeeze(0).transpose(0, 1)
            dv_ref[s0:s1] = dvs.squeeze(0).transpose(0, 1)
        _assert_close_bf16(out, o_ref, atol=3e-2, rtol=3e-2)
        _assert_close_bf16(q.grad, dq_ref, atol=5e-2, rtol=5e-2)
        _assert_close_bf16(k.grad, dk_ref, atol=5e-2, rtol=5e-2)
        _assert_close_bf16(v.grad, dv_ref, atol=5e-2, rtol=5e-2)


# ---------------------------------------------------------------------------
# fp8 e4m3 MX GEMM (gemm_fp8.hip)
# ---------------------------------------------------------------------------
def test_fp8mx_probe_layout():
    torch.manual_seed(30)
    a = (torch.randn(16, 128) * 0.5).to(torch.float8_e4m3fn)
    bt = (torch.randn(16, 128) * 0.5).to(torch.float8_e4m3fn)
    one = 0x7F7F7F7F
    C = _ext.get_ext()
    out = C.mfma_probe_fp8mx(a.view(torch.uint8).cuda(),
                             bt.view(torch.uint8).cuda(), one, one).cpu()
    ref = a.float() @ bt.float().t()
    _gemm_rel_ok(out, ref, tol=1e-3)


@pytest.mark.parametrize("m,n,k", [(512, 512, 512), (777, 300, 256),
                                   (2048, 1024, 4096)])
def test_gemm_fp8_numerics(m, n, k):
    torch.manual_seed(31)
    C = _ext.get_ext()
    a = (torch.randn(m, k, device=DEV) * 0.3).to(torch.float8_e4m3fn)
    bt = (torch.randn(n, k, device=DEV) * 0.3).to(torch.float8_e4m3fn)
    out = C.gemm_fp8_nt(a, bt, 0.731)
    ref = 0.731 * (a.float() @ bt.float().t())
    _gemm_rel_ok(out, ref, tol=2e-2)
    # bias epilogue
    bias = _bf(torch.randn(n, device=DEV))
    outb = C.gemm_fp8_nt(a, bt, 1.0, bias)
    _gemm_rel_ok(outb, a.float() @ bt.float().t() + bias.float(), tol=2e-2)


def test_fp8_matmul_roundtrip():
    from paddle_amd.incubate import fp8
    torch.manual_seed(32)
    x = _bf(torch.randn(512, 512, device=DEV))
    w = _bf(torch.randn(512, 384, device=DEV) * 0.05)
    y = fp8.fp8_matmul(x, w)
    ref = x.float() @ w.float()
    _gemm_rel_ok(y, ref, tol=6e-2)  # fp8 quantization error budget


# ---------------------------------------------------------------------------
# MoE routing kernels (moe.hip)
# ---------------------------------------------------------------------------
def test_moe_gate_topk_kernel():
    torch.manual_seed(40)
    C = _ext.get_ext()
    T, E, K = 4096, 64, 2
    logits = torch.randn(T, E, device=DEV)
    topv, topi, me, ce = C.moe_gate_topk(logits, K)
    p = torch.softmax(logits, -1)
    rv, ri = p.topk(K, -1)
    torch.testing.assert_close(topv, rv, atol=1e-5, rtol=1e-5)
    assert (topi.long() == ri).float().mean() > 0.999  # ties may differ
    torch.testing.assert_close(me, p.mean(0), atol=1e-4, rtol=1e-4)
    ce_ref = torch.bincount(ri[:, 0], minlength=E).float() / T
    torch.testing.assert_close(ce, ce_ref, atol=1e-4, rtol=1e-4)


def test_moe_assign_slots_kernel():
    torch.manual_seed(41)
    C = _ext.get_ext()
    T, E, K, cap = 2048, 16, 2, 100
    topi = torch.randint(0, E, (T, K), device=DEV, dtype=torch.int32)
    slot, counts = C.moe_assign_slots(topi, E, cap)
    # reference: deterministic token-order positions per (expert, k)
    ref = torch.full((T, K), -1, dtype=torch.long)
    cnt = torch.zeros(E, dtype=torch.long)
    ti = topi.cpu()
    pos_ctr = {}
    for kk in range(K):
        pc = [0] * E
        for t in range(T):
            e = int(ti[t, kk])
            if pc[e] < cap:
                ref[t, kk] = e * cap + pc[e]
            pc[e] += 1
            cnt[e] += 1
    assert (slot.cpu().long() == ref).all()
    assert (counts.cpu().long() == cnt).all()


def test_moe_gate_gradients_gpu():
    """Fused gate backward vs torch autograd reference."""
    torch.manual_seed(42)
    from paddle_amd.models.moe import _FusedGate
    T, E, K = 512, 32, 2
    logits = torch.randn(T, E, device=DEV, requires_grad=True)
    topv, topi, aux = _FusedGate.apply(logits, K, E)
    loss = topv.square().sum() + 3.0 * aux
    loss.backward()
    lr = logits.detach().clone().requires_grad_(True)
    p = torch.softmax(lr, -1)
    rv, ri = p.topk(K, -1)
    me = p.mean(0)
    ce = torch.bincount(ri[:, 0], minlength=E).float() / T
    (rv.square().sum() + 3.0 * (me * ce).sum() * E).backward()
    torch.testing.assert_close(logits.grad, lr.grad, atol=1e-4, rtol=1e-4)


# ---------------------------------------------------------------------------
# skinny decode GEMM (decode_gemm.hip)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("m,k,n", [(1, 4096, 4096), (16, 4096, 16384),
                                   (32, 16384, 4096), (16, 4096, 50304),
                                   (7, 2048, 1100)])
def test_decode_gemm_numerics(m, k, n):
    torch.manual_seed(50)
    C = _ext.get_ext()
    x = _bf(torch.randn(m, k, device=DEV))
    w = _bf(torch.randn(k, n, device=DEV) * 0.02)
    bias = _bf(torch.randn(n, device=DEV))
    y = C.decode_gemm(x, w, bias)
    ref = x.float() @ w.float() + bias.float()
    _gemm_rel_ok(y, ref, tol=2e-2)
    y2 = C.decode_gemm(x, w, None)
    _gemm_rel_ok(y2, x.float() @ w.float(), tol=2e-2)


@pytest.mark.parametrize("m,k,n", [(32, 4096, 4096), (16, 512, 1024),
                                   (5, 256, 512), (32, 4160, 12288)])
def test_decode_gemm_mfma_numerics(m, k, n):
    torch.manual_seed(4)
    C = _ext.get_ext()
    x = _bf(torch.randn(m, k, device=DEV))
    w = _bf(torch.randn(k, n, device=DEV) * 0.02)
    bias = _bf(torch.randn(n, device=DEV))
    y = C.decode_gemm_mfma(x, w, bias)
    ref = x.float() @ w.float() + bias.float()
    _gemm_rel_ok(y, ref, tol=2e-2)
    y2 = C.decode_gemm_mfma(x, w, None)
    _gemm_rel_ok(y2, x.float() @ w.float(), tol=2e-2)
