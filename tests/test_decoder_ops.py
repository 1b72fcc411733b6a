"""Fused decoder-op parity on CPU (masked_mha / block_mha /
fused_multi_transformer); GPU numerics live in test_kernels_gpu.py."""
import math

import torch

import paddle_amd  # noqa: F401
from paddle_amd.incubate.nn import functional as inf
from paddle_amd.ops.functional import _sdpa_ref


def test_masked_multihead_attention_cpu():
    torch.manual_seed(0)
    B, H, MS, D = 2, 4, 16, 64
    cache = torch.zeros(2, B, H, MS, D)
    cache[0, :, :, :6] = torch.randn(B, H, 6, D)
    cache[1, :, :, :6] = torch.randn(B, H, 6, D)
    x = torch.randn(B, 3 * H * D)
    lens = torch.tensor([3, 5], dtype=torch.int32)
    out, cache2 = inf.masked_multihead_attention(x, cache, sequence_lengths=lens)
    q = x.reshape(B, 3, H, D)[0, 0]
    k, v = cache2[0, 0, :, :4], cache2[1, 0, :, :4]
    s = torch.einsum("hd,hsd->hs", q, k) / math.sqrt(D)
    ref = torch.einsum("hs,hsd->hd", torch.softmax(s.float(), -1), v.float())
    torch.testing.assert_close(out[0].reshape(H, D), ref, atol=1e-4, rtol=1e-4)
    # the new token landed in the cache at position lens[b]
    torch.testing.assert_close(cache2[0][0, :, 3], x.reshape(B, 3, H, D)[0, 1])


def test_block_multihead_attention_cpu():
    torch.manual_seed(1)
    B, H, D, bs, maxblk = 2, 2, 64, 4, 6
    kc = torch.zeros(16, bs, H, D)
    vc = torch.zeros(16, bs, H, D)
    bt = torch.arange(B * maxblk, dtype=torch.int32).reshape(B, maxblk)
    enc = torch.tensor([7, 0])
    dec = torch.tensor([0, 5])
    this = torch.tensor([7, 1])
    hist_k = torch.randn(5, H, D)
    hist_v = torch.randn(5, H, D)
    for p in range(5):
        kc[int(bt[1, p // bs]), p % bs] = hist_k[p]
        vc[int(bt[1, p // bs]), p % bs] = hist_v[p]
    qkv = torch.randn(8, 3 * H * D)
    out, _, kc, vc = inf.block_multihead_attention(qkv, kc, vc, enc, dec, this,
                                                   block_tables=bt, block_size=bs)
    q3 = qkv.reshape(8, 3, H, D)
    kk = torch.cat([hist_k, q3[7, 1].unsqueeze(0)])
    vv = torch.cat([hist_v, q3[7, 2].unsqueeze(0)])
    s = torch.einsum("hd,shd->hs", q3[7, 0], kk) / math.sqrt(D)
    ref = torch.einsum("hs,shd->hd", torch.softmax(s.float(), -1), vv.float())
    torch.testing.assert_close(out[7].reshape(H, D), ref, atol=1e-4, rtol=1e-4)
    qs = q3[:7, 0].transpose(0, 1).unsqueeze(0)
    ks = q3[:7, 1].transpose(0, 1).unsqueeze(0)
    vs = q3[:7, 2].transpose(0, 1).unsqueeze(0)
    refp, _ = _sdpa_ref(qs.float(), ks.float(), vs.float(), 1 / math.sqrt(D), True)
    torch.testing.assert_close(out[:7].reshape(7, H, D),
                               refp.squeeze(0).transpose(0, 1), atol=1e-4, rtol=1e-4)


def test_fused_multi_transformer_decode_matches_context():
    torch.manual_seed(2)
    B, S, d, H, D, L = 2, 8, 64, 4, 16, 2
    mk = lambda *s: torch.randn(*s) * 0.05
    args = dict(
        ln_scales=[torch.ones(d)] * L, ln_biases=[torch.zeros(d)] * L,
        qkv_weights=[mk(3 * H * D, d) for _ in range(L)],
        qkv_biases=[torch.zeros(3 * H * D)] * L,
        linear_weights=[mk(H * D, d) for _ in range(L)],
        linear_biases=[torch.zeros(d)] * L,
        ffn_ln_scales=[torch.ones(d)] * L, ffn_ln_biases=[torch.zeros(d)] * L,
        ffn1_weights=[mk(d, 4 * d) for _ in range(L)],
        ffn1_biases=[torch.zeros(4 * d)] * L,
        ffn2_weights=[mk(4 * d, d) for _ in range(L)],
        ffn2_biases=[torch.zeros(d)] * L,
    )
    caches = [torch.zeros(2, B, H, 32, D) for _ in range(L)]
    x = torch.randn(B, S, d)
    out, caches = inf.fused_multi_transformer(x, cache_kvs=caches, **args)
    x1 = torch.randn(B, 1, d)
    out1, _ = inf.fused_multi_transformer(x1, cache_kvs=caches, time_step=S, **args)
    caches2 = [torch.zeros(2, B, H, 32, D) for _ in range(L)]
    out_full, _ = inf.fused_multi_transformer(torch.cat([x, x1], 1),
                                              cache_kvs=caches2, **args)
    torch.testing.assert_close(out_full[:, -1:], out1, atol=1e-3, rtol=1e-3)


def test_variable_length_memory_efficient_attention_cpu():
    torch.manual_seed(3)
    B, H, S, D = 2, 2, 16, 32
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    lens = torch.tensor([10, 16])
    out = inf.variable_length_memory_efficient_attention(q, k, v, seq_lens=lens)
    for i, L in enumerate(lens.tolist()):
        ref, _ = _sdpa_ref(q[i:i+1, :, :L].float(), k[i:i+1, :, :L].float(),
                           v[i:i+1, :, :L].float(), 1 / math.sqrt(D), False)
        torch.testing.assert_close(out[i, :, :L], ref.squeeze(0), atol=1e-4,
                                   rtol=1e-4)


def test_blha_get_max_len():
    import torch
    from paddle_amd.incubate.nn import functional as F
    enc = torch.tensor([3, 9, 0], dtype=torch.int32)
    dec = torch.tensor([5, 2, 7], dtype=torch.int32)
    me, md = F.blha_get_max_len(enc, dec, torch.tensor(3))
    assert int(me) == 9 and int(md) == 7 and me.shape == (1,)
