"""paddle.incubate.nn.functional parity: fused functional ops.

Reference signatures: SURVEY.md A.7 (fused_rms_norm, fused_rotary_
position_embedding, fused_dropout_add, swiglu, fused_bias_act...).
"""
from __future__ import annotations

import torch
import math

from ....ops import functional as hot
from ....ops.functional import (  # noqa: F401
    fused_rotary_position_embedding,
    swiglu,
)


def fused_rms_norm(x, norm_weight, norm_bias=None, epsilon=1e-6, begin_norm_axis=-1,
                   bias=None, residual=None, quant_scale=-1, **kw):
    out = hot.fused_rms_norm(x, norm_weight, residual=residual, epsilon=epsilon)
    if residual is not None:
        return out  # (y, residual_out)
    return out, None


def fused_layer_norm(x, norm_weight, norm_bias, epsilon=1e-5, begin_norm_axis=-1,
                     bias=None, residual=None, **kw):
    if residual is not None:
        x = x + residual
        res_out = x
        y = hot.layer_norm(x, norm_weight, norm_bias, epsilon)
        return y, res_out
    return hot.layer_norm(x, norm_weight, norm_bias, epsilon), None


def fused_dropout_add(x, y, p=0.5, training=True, mode="upscale_in_train", name=None):
    return hot.dropout_add(x, y, p, training)


def fused_bias_act(x, bias=None, act_method="gelu", **kw):
    if act_method == "gelu":
        return hot.bias_gelu(x, bias)
    if act_method == "swiglu":
        if bias is not None:
            x = x + bias
        return hot.swiglu(x)
    raise ValueError(act_method)


def fused_linear(x, weight, bias=None, transpose_weight=False, name=None):
    w = weight.t() if transpose_weight else weight
    out = torch.matmul(x, w)
    if bias is not None:
        out = out + bias
    return out


def fused_linear_activation(x, y, bias, trans_x=False, trans_y=False, activation="gelu"):
    if trans_x:
        x = x.transpose(-1, -2)
    w = y.t() if trans_y else y
    out = torch.matmul(x, w)
    if activation == "gelu":
        return hot.bias_gelu(out, bias)
    out = out + bias
    if activation == "relu":
        out = torch.relu(out)
    return out


# fused functional forms (reference: incubate/nn/functional/__init__.py)
import torch as _t


def fused_matmul_bias(x, y, bias=None, transpose_x=False, transpose_y=False,
                      name=None):
    """GEMM + bias in one hipBLASLt call (torch.addmm epilogue path)."""
    a = x.t() if transpose_x else x
    b = y.t() if transpose_y else y
    if bias is not None and a.dim() == 2:
        return _t.addmm(bias, a, b)
    out = a @ b
    return out + bias if bias is not None else out


def fused_multi_head_attention(x, qkv_weight, linear_weight, pre_layer_norm=False,
                               pre_ln_scale=None, pre_ln_bias=None, ln_scale=None,
                               ln_bias=None, pre_ln_epsilon=1e-5, qkv_bias=None,
                               linear_bias=None, cache_kv=None, attn_mask=None,
                               dropout_rate=0.0, attn_dropout_rate=0.0,
                               ln_epsilon=1e-5, training=True, mode='upscale_in_train',
                               ring_id=-1, add_residual=True, num_heads=None,
                               transpose_qkv_wb=False, name=None):
    """Functional form of FusedMultiHeadAttention (reference:
    fused_attention_kernel.cu flow: [preLN ->] qkv -> FMHA -> proj ->
    [dropout ->] residual [-> LN])."""
    from ....ops import functional as hot
    residual = x
    h = x
    if pre_layer_norm:
        h = hot.layer_norm(h, pre_ln_scale, pre_ln_bias, pre_ln_epsilon)
    b, s, d = h.shape
    if transpose_qkv_wb:
        qkv = h @ qkv_weight            # [d, 3d]
        if qkv_bias is not None:
            qkv = qkv + qkv_bias
        nh = num_heads
        qkv = qkv.reshape(b, s, 3, nh, d // nh)
    else:
        nh = qkv_weight.shape[1]
        qkv = _t.einsum("bsd,thnd->bsthn" if qkv_weight.dim() == 4 else "bsd,dt->bst",
                        h, qkv_weight)
        if qkv.dim() == 5:
            pass
        if qkv_bias is not None:
            qkv = qkv + qkv_bias
        qkv = qkv.reshape(b, s, 3, nh, d // nh)
    att = hot.qkv_flash_attention(qkv.contiguous(), causal=False)
    att = att.reshape(b, s, d)
    out = att @ linear_weight
    if linear_bias is not None:
        out = out + linear_bias
    if dropout_rate and training:
        out = _t.nn.functional.dropout(out, dropout_rate)
    if add_residual:
        out = out + residual
    if not pre_layer_norm and ln_scale is not None:
        from ....ops import functional as hot2
        out = hot2.layer_norm(out, ln_scale, ln_bias, ln_epsilon)
    return out


def fused_feedforward(x, linear1_weight, linear2_weight, linear1_bias=None,
                      linear2_bias=None, ln1_scale=None, ln1_bias=None,
                      ln2_scale=None, ln2_bias=None, dropout1_rate=0.5,
                      dropout2_rate=0.5, activation="relu", ln1_epsilon=1e-5,
                      ln2_epsilon=1e-5, pre_layer_norm=False, training=True,
                      mode='upscale_in_train', ring_id=-1, add_residual=True,
                      name=None):
    """Functional FusedFeedForward (reference: fused_feedforward_kernel.cu:
    [preLN ->] GEMM1 -> act(+dropout) -> GEMM2 (+dropout) -> residual
    [-> LN])."""
    from ....ops import functional as hot
    residual = x
    h = x
    if pre_layer_norm:
        h = hot.layer_norm(h, ln1_scale, ln1_bias, ln1_epsilon)
    h = h @ linear1_weight
    if activation == "gelu":
        h = hot.bias_gelu(h, linear1_bias)
    else:
        if linear1_bias is not None:
            h = h + linear1_bias
        h = _t.relu(h)
    if dropout1_rate and training:
        h = _t.nn.functional.dropout(h, dropout1_rate)
    h = h @ linear2_weight
    if linear2_bias is not None:
        h = h + linear2_bias
    if dropout2_rate and training:
        h = _t.nn.functional.dropout(h, dropout2_rate)
    out = h + residual if add_residual else h
    if not pre_layer_norm and ln2_scale is not None:
        out = hot.layer_norm(out, ln2_scale, ln2_bias, ln2_epsilon)
    return out


def fused_bias_dropout_residual_layer_norm(x, residual, bias=None,
                                           ln_scale=None, ln_bias=None,
                                           dropout_rate=0.5, ln_epsilon=1e-5,
                                           training=True,
                                           mode='upscale_in_train', name=None):
    from ....ops import functional as hot
    h = x + bias if bias is not None else x
    h = hot.dropout_add(h, residual, dropout_rate, training)
    return hot.layer_norm(h, ln_scale, ln_bias, ln_epsilon)


def masked_multihead_attention(x, cache_kv=None, bias=None, src_mask=None,
                               cum_offsets=None, sequence_lengths=None,
                               rotary_tensor=None, beam_cache_offset=None,
                               qkv_out_scale=None, out_shift=None,
                               out_smooth=None, seq_len=1, rotary_emb_dims=0,
                               use_neox_rotary_style=False,
                               compute_dtype="default", out_scale=-1,
                               quant_round_type=1, quant_max_bound=127.0,
                               quant_min_bound=-127.0):
    """Single-token decode attention over a dense KV cache.

    Parity: paddle/phi/kernels/fusion/gpu/masked_multihead_attention_kernel.cu
    (python incubate/nn/functional/masked_multihead_attention.py).
    x: [B, 3*H*D] fused qkv for ONE step; cache_kv [2, B, H, max_seq, D]
    (written in place at the current length).  Returns (out [B, H*D],
    cache_kv).  The gfx950 path runs csrc/kernels/decode_attn.hip with
    dense-cache strides; int8/quant args are unsupported.
    """
    import torch as _t
    from .... import _ext
    if qkv_out_scale is not None or out_shift is not None or out_smooth is not None \
            or out_scale not in (-1, -1.0):
        raise NotImplementedError("masked_multihead_attention: quant args unsupported")
    assert cache_kv is not None and cache_kv.dim() == 5
    two, B, H, MS, D = cache_kv.shape
    qkv = x if bias is None else x + bias
    qkv = qkv.reshape(B, 3, H, D)
    q, k, v = qkv[:, 0], qkv[:, 1], qkv[:, 2]
    if sequence_lengths is not None:
        lens = sequence_lengths.reshape(-1).to(_t.int32)
    elif src_mask is not None:
        # src_mask [B, 1, 1, S]: finite entries mark attendable positions
        lens = (src_mask.reshape(B, -1) > -1e4).sum(-1).to(_t.int32) - 1
        lens = lens.clamp(min=0)
    else:
        lens = _t.zeros(B, dtype=_t.int32, device=x.device)
    if rotary_emb_dims > 0 and rotary_tensor is not None:
        # rotary_tensor [2, B, 1, seq, D] (cos|sin) at the current position
        cos = rotary_tensor[0].reshape(B, -1, D)[_t.arange(B), lens.long()]
        sin = rotary_tensor[1].reshape(B, -1, D)[_t.arange(B), lens.long()]
        cos = cos.unsqueeze(1)
        sin = sin.unsqueeze(1)

        def _rot(t):
            if use_neox_rotary_style:
                h1, h2 = t.chunk(2, -1)
                rot = _t.cat([-h2, h1], -1)
            else:
                e, o = t[..., 0::2], t[..., 1::2]
                rot = _t.stack([-o, e], -1).reshape(t.shape)
            return t * cos + rot * sin
        q = _rot(q)
        k = _rot(k)
    # append this step's k/v at position lens[b]
    bi = _t.arange(B, device=x.device)
    cache_kv[0][bi, :, lens.long()] = k.to(cache_kv.dtype)
    cache_kv[1][bi, :, lens.long()] = v.to(cache_kv.dtype)
    attn_lens = (lens + 1).contiguous()
    if (_ext.use_native(x) and x.dtype == _t.bfloat16 and D in (64, 128)
            and src_mask is None):
        C = _ext.get_ext()
        bt = bi.to(_t.int32).reshape(B, 1).contiguous()
        out = C.decode_attention(q.contiguous(), cache_kv[0], cache_kv[1], bt,
                                 attn_lens, 1.0 / math.sqrt(D),
                                 H * MS * D, D, MS * D, MS)
    else:
        # exact torch path (also covers additive src_mask)
        s = _t.einsum("bhd,bhsd->bhs", q.float(), cache_kv[0].float())
        s = s / math.sqrt(D)
        pos = _t.arange(MS, device=x.device).reshape(1, 1, MS)
        s = s.masked_fill(pos >= attn_lens.reshape(B, 1, 1), float("-inf"))
        if src_mask is not None:
            s = s + src_mask.reshape(B, 1, -1)[:, :, :MS].float()
        p = _t.softmax(s, -1)
        out = _t.einsum("bhs,bhsd->bhd", p, cache_kv[1].float()).to(x.dtype)
    return out.reshape(B, H * D), cache_kv


def block_multihead_attention(qkv, key_cache, value_cache, seq_lens_encoder,
                              seq_lens_decoder, seq_lens_this_time,
                              padding_offsets=None, cum_offsets=None,
                              cu_seqlens_q=None, cu_seqlens_k=None,
                              block_tables=None, pre_key_cache=None,
                              pre_value_cache=None, cache_k_quant_scales=None,
                              cache_v_quant_scales=None,
                              cache_k_dequant_scales=None,
                              cache_v_dequant_scales=None, qkv_out_scale=None,
                              qkv_bias=None, out_shift=None, out_smooth=None,
                              max_enc_len_this_time=None,
                              max_dec_len_this_time=None, rope_emb=None,
                              mask=None, tgt_mask=None, max_input_length=0,
                              block_size=64, use_neox_style=False,
                              use_dynamic_cachekv_quant=False,
                              quant_round_type=1, quant_max_bound=127.0,
                              quant_min_bound=-127.0, out_scale=-1,
                              compute_dtype="default"):
    """Paged ("block") KV-cache attention: prefill sequences go through the
    single-launch varlen flash kernel and write their KV into the paged
    cache; decode sequences (1 token) run decode_attn.hip.

    Parity: paddle/phi/kernels/fusion/gpu/block_multi_head_attention_kernel.cu
    (block_attn.h).  qkv: packed [total_tokens, 3*H*D]; key/value_cache
    [nblocks, block_size, HKV, D]; block_tables [B, max_blocks].
    Quant / pre-cache args are unsupported.  Returns
    (fmha_out [total, H*D], qkv, key_cache, value_cache).
    """
    import torch as _t
    from ....ops.functional import flash_attn_varlen_func
    if cache_k_quant_scales is not None or pre_key_cache is not None:
        raise NotImplementedError("block_multihead_attention: quant/pre-cache unsupported")
    from .... import _ext
    nblocks, bs, HKV, D = key_cache.shape
    B = seq_lens_this_time.shape[0]
    total = qkv.shape[0]
    H = qkv.shape[-1] // (3 * D)
    if qkv_bias is not None:
        qkv = qkv + qkv_bias
    q3 = qkv.reshape(total, 3, H, D)
    q, k, v = q3[:, 0], q3[:, 1], q3[:, 2]
    enc = seq_lens_encoder.reshape(-1).tolist()
    dec = seq_lens_decoder.reshape(-1).tolist()
    this = seq_lens_this_time.reshape(-1).tolist()
    out = _t.empty(total, H, D, dtype=qkv.dtype, device=qkv.device)
    # token offsets per sequence within the packed batch
    offs = [0]
    for t_ in this:
        offs.append(offs[-1] + int(t_))

    def _write_cache(bidx, start_pos, kk, vv):
        for j in range(kk.shape[0]):
            pos = start_pos + j
            phys = int(block_tables[bidx, pos // bs])
            key_cache[phys, pos % bs] = kk[j]
            value_cache[phys, pos % bs] = vv[j]

    # ---- prefill (encoder) sequences: one varlen FA launch ----------------
    pre_idx = [i for i in range(B) if enc[i] > 0]
    if pre_idx:
        qs = _t.cat([q[offs[i]:offs[i + 1]] for i in pre_idx])
        ks = _t.cat([k[offs[i]:offs[i + 1]] for i in pre_idx])
        vs = _t.cat([v[offs[i]:offs[i + 1]] for i in pre_idx])
        cu = _t.tensor([0] + list(_t.cumsum(_t.tensor([this[i] for i in pre_idx]), 0)),
                       dtype=_t.int32, device=qkv.device)
        o = flash_attn_varlen_func(qs, ks, vs, cu, cu, max(this), max(this),
                                   causal=True)
        p0 = 0
        for i in pre_idx:
            n = this[i]
            out[offs[i]:offs[i + 1]] = o[p0:p0 + n]
            _write_cache(i, 0, k[offs[i]:offs[i + 1]], v[offs[i]:offs[i + 1]])
            p0 += n
    # ---- decode sequences: append + paged decode kernel -------------------
    dec_idx = [i for i in range(B) if enc[i] == 0 and this[i] > 0]
    if dec_idx:
        for i in dec_idx:
            _write_cache(i, dec[i], k[offs[i]:offs[i + 1]], v[offs[i]:offs[i + 1]])
        qd = _t.stack([q[offs[i]] for i in dec_idx])
        lens = _t.tensor([dec[i] + 1 for i in dec_idx], dtype=_t.int32,
                         device=qkv.device)
        bt = block_tables[dec_idx].to(_t.int32).contiguous()
        if _ext.use_native(qkv) and qkv.dtype == _t.bfloat16 and D in (64, 128):
            C = _ext.get_ext()
            od = C.decode_attention(qd.contiguous(), key_cache, value_cache,
                                    bt, lens, 1.0 / math.sqrt(D))
        else:
            od = _t.empty_like(qd)
            for j, i in enumerate(dec_idx):
                L = dec[i] + 1
                kk = _t.stack([key_cache[int(block_tables[i, p // bs]), p % bs]
                               for p in range(L)])
                vv = _t.stack([value_cache[int(block_tables[i, p // bs]), p % bs]
                               for p in range(L)])
                sc = _t.einsum("hd,shd->hs", qd[j].float(), kk.float()) / math.sqrt(D)
                pr = _t.softmax(sc, -1)
                od[j] = _t.einsum("hs,shd->hd", pr, vv.float()).to(qkv.dtype)
        for j, i in enumerate(dec_idx):
            out[offs[i]] = od[j]
    return out.reshape(total, H * D), qkv, key_cache, value_cache


def fused_multi_transformer(x, ln_scales, ln_biases, qkv_weights, qkv_biases,
                            linear_weights, linear_biases, ffn_ln_scales,
                            ffn_ln_biases, ffn1_weights, ffn1_biases,
                            ffn2_weights, ffn2_biases, pre_layer_norm=True,
                            epsilon=1e-5, cache_kvs=None, pre_caches=None,
                            rotary_embs=None, beam_offset=None, time_step=None,
                            seq_lens=None, attn_mask=None, dropout_rate=0.0,
                            activation="gelu", training=False,
                            mode="upscale_in_train", trans_qkvw=True,
                            ring_id=-1, name=None):
    """Whole-decoder-stack fused op (inference).

    Parity: paddle/phi/kernels/fusion/gpu/fused_multi_transformer_kernel.cu
    (python incubate/nn/functional/fused_transformer.py).  Context phase
    (time_step None): causal flash attention over [B, S, d], caches filled.
    Generation phase (time_step set): one-token decode through
    masked_multihead_attention on the dense caches.  Returns
    (out, cache_kvs).
    """
    import torch as _t
    from ....ops import functional as hot
    num_layers = len(qkv_weights)
    B, S, dmodel = x.shape
    H3 = qkv_weights[0].shape[0] if trans_qkvw else qkv_weights[0].shape[-1]
    decode = time_step is not None
    out = x
    for li in range(num_layers):
        residual = out
        h = hot.layer_norm(out, ln_scales[li], ln_biases[li], epsilon) \
            if pre_layer_norm else out
        wq = qkv_weights[li]
        if trans_qkvw:          # [3*H*D, d]
            qkv = h.reshape(-1, dmodel) @ wq.t()
        else:
            qkv = h.reshape(-1, dmodel) @ wq.reshape(dmodel, -1)
        if qkv_biases is not None and qkv_biases[li] is not None:
            qkv = qkv + qkv_biases[li].reshape(-1)
        cache = cache_kvs[li] if cache_kvs is not None else None
        if decode:
            assert S == 1
            lens = seq_lens if seq_lens is not None else \
                _t.full((B,), int(time_step), dtype=_t.int32, device=x.device)
            attn, _ = masked_multihead_attention(
                qkv.reshape(B, -1), cache, sequence_lengths=lens,
                rotary_tensor=rotary_embs[li] if isinstance(rotary_embs, (list, tuple)) else rotary_embs,
                rotary_emb_dims=1 if rotary_embs is not None else 0)
            attn = attn.reshape(B, 1, -1)
        else:
            nH = cache.shape[2] if cache is not None else H3 // (3 * 128)
            D = (qkv.shape[-1] // 3) // nH
            q3 = qkv.reshape(B, S, 3, nH, D)
            q, k, v = q3[:, :, 0], q3[:, :, 1], q3[:, :, 2]
            o, _ = hot.flash_attention(q, k, v, causal=True)
            attn = o.reshape(B, S, nH * D)
            if cache is not None:
                cache[0][:, :, :S] = k.transpose(1, 2)
                cache[1][:, :, :S] = v.transpose(1, 2)
        lin = attn.reshape(-1, attn.shape[-1]) @ linear_weights[li].reshape(
            attn.shape[-1], dmodel)
        if linear_biases is not None and linear_biases[li] is not None:
            lin = lin + linear_biases[li]
        h = residual + lin.reshape(B, S, dmodel)
        residual2 = h
        h2 = hot.layer_norm(h, ffn_ln_scales[li], ffn_ln_biases[li], epsilon) \
            if pre_layer_norm else h
        f1 = h2.reshape(-1, dmodel) @ ffn1_weights[li].reshape(dmodel, -1)
        if ffn1_biases is not None and ffn1_biases[li] is not None:
            f1 = f1 + ffn1_biases[li]
        f1 = _t.nn.functional.gelu(f1) if activation == "gelu" \
            else _t.nn.functional.relu(f1)
        f2 = f1 @ ffn2_weights[li].reshape(f1.shape[-1], dmodel)
        if ffn2_biases is not None and ffn2_biases[li] is not None:
            f2 = f2 + ffn2_biases[li]
        out = residual2 + f2.reshape(B, S, dmodel)
        if ring_id >= 0:
            from ....distributed import collective as _c
            _c.all_reduce(out)
    return out, cache_kvs


def fused_moe(x, gate_weight, expert_weights1, expert_weights2, **kwargs):
    raise NotImplementedError(
        "fused_moe (cutlass inference kernel): use models.moe.MoELayer "
        "with grouped experts")


def variable_length_memory_efficient_attention(query, key, value, seq_lens=None,
                                               kv_seq_lens=None, mask=None,
                                               scale=None, causal=False,
                                               pre_cache_length=0):
    """Padded-batch ragged attention (reference:
    fusion/cutlass/variable_length_memory_efficient_attention.cu).
    query/key/value: [B, H, S, D] padded; seq_lens [B] valid lengths.
    Packs to the single-launch varlen flash kernel."""
    import torch as _t
    from ....ops.functional import flash_attn_varlen_func, flash_attention
    if pre_cache_length:
        raise NotImplementedError("pre_cache unsupported")
    B, H, S, D = query.shape
    if seq_lens is None:
        out, _ = flash_attention(query.transpose(1, 2), key.transpose(1, 2),
                                 value.transpose(1, 2), causal=causal,
                                 scale=scale, attn_mask=mask)
        return out.transpose(1, 2)
    lens = seq_lens.reshape(-1).tolist()
    klens = kv_seq_lens.reshape(-1).tolist() if kv_seq_lens is not None else lens
    qp = _t.cat([query[i, :, :lens[i]].transpose(0, 1) for i in range(B)])
    kp = _t.cat([key[i, :, :klens[i]].transpose(0, 1) for i in range(B)])
    vp = _t.cat([value[i, :, :klens[i]].transpose(0, 1) for i in range(B)])
    cu_q = _t.tensor([0] + list(_t.cumsum(_t.tensor(lens), 0)), dtype=_t.int32,
                     device=query.device)
    cu_k = _t.tensor([0] + list(_t.cumsum(_t.tensor(klens), 0)), dtype=_t.int32,
                     device=query.device)
    op = flash_attn_varlen_func(qp.contiguous(), kp.contiguous(), vp.contiguous(),
                                cu_q, cu_k, max(lens), max(klens), scale=scale,
                                causal=causal)
    out = _t.zeros_like(query)
    for i in range(B):
        s0, s1 = int(cu_q[i]), int(cu_q[i + 1])
        out[i, :, :lens[i]] = op[s0:s1].transpose(0, 1)
    return out


def blha_get_max_len(seq_lens_encoder, seq_lens_decoder, batch_size):
    """Max encoder/decoder sequence lengths for the current step, used
    ahead of block_multihead_attention (reference:
    incubate/nn/functional/blha_get_max_len.py:26 -> fused
    BlhaGetMaxLen kernel; a two-max reduction here)."""
    import torch
    max_enc = seq_lens_encoder.max().reshape([1])
    max_dec = seq_lens_decoder.max().reshape([1])
    return max_enc, max_dec
