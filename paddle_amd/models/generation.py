"""Autoregressive generation with a paged KV cache (serving path).

Pairs the prefill flash-attention kernel with the paged decode kernel
(csrc/kernels/decode_attn.hip) -- the reference's
fused_multi_transformer / block_multihead_attention role.
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from ..ops import functional as hot


class PagedKVCache:
    """Per-layer paged cache: k/v [nblocks, block_size, HKV, D]."""

    def __init__(self, num_layers, num_blocks, block_size, num_kv_heads, head_dim,
                 batch, max_seq, device, dtype=torch.bfloat16):
        self.bs = block_size
        self.num_layers = num_layers
        self.k = [torch.zeros(num_blocks, block_size, num_kv_heads, head_dim,
                              device=device, dtype=dtype) for _ in range(num_layers)]
        self.v = [torch.zeros_like(self.k[0]) for _ in range(num_layers)]
        max_blocks = (max_seq + block_size - 1) // block_size
        assert num_blocks >= batch * max_blocks, "cache too small"
        # static block allocation: seq b owns blocks [b*max_blocks, ...)
        self.block_table = (torch.arange(batch * max_blocks, device=device)
                            .reshape(batch, max_blocks).int())
        self.seq_lens = torch.zeros(batch, device=device, dtype=torch.int32)

    def append(self, layer, k_new, v_new, start_pos):
        """k_new/v_new: [B, S_new, HKV, D]; positions start_pos..start_pos+S_new
        (vectorized scatter: one advanced-indexing write per layer)"""
        B, S_new, HKV, D = k_new.shape
        pos = torch.arange(start_pos, start_pos + S_new,
                           device=k_new.device)               # [S]
        blks = self.block_table.long()[:, pos // self.bs]     # [B, S]
        offs = (pos % self.bs).unsqueeze(0).expand(B, S_new)  # [B, S]
        self.k[layer][blks, offs] = k_new
        self.v[layer][blks, offs] = v_new

    def advance(self, n):
        self.seq_lens += n


@torch.no_grad()
def generate_gpt(model, input_ids, max_new_tokens=16, cache_blocks=None,
                 block_size=16, greedy=True):
    """Greedy/sampled generation for GPTForPretraining with paged KV."""
    model.eval()
    cfg = model.cfg
    B, S0 = input_ids.shape
    dev = input_ids.device
    H = cfg.num_heads
    D = cfg.hidden_size // H
    max_seq = S0 + max_new_tokens
    nblocks = cache_blocks or (B * ((max_seq + block_size - 1) // block_size))
    cache = PagedKVCache(cfg.num_layers, nblocks, block_size, H, D, B, max_seq, dev,
                         dtype=model.lm_head.weight.dtype)

    def layer_qkv(layer, x):
        b, s, _ = x.shape
        qkv = layer.attn.qkv_proj(x).reshape(b, s, 3, H, D)
        return qkv.unbind(2)  # q,k,v [b,s,H,D]

    # ---- prefill ----------------------------------------------------------
    x = model.gpt.embeddings(input_ids)
    for li, layer in enumerate(model.gpt.layers):
        h = layer.ln1(x)
        q, k, v = layer_qkv(layer, h)
        cache.append(li, k, v, 0)
        att, _ = hot.flash_attention(q, k, v, causal=True)
        att = layer.attn.out_proj(att.reshape(att.shape[0], att.shape[1], -1))
        x = x + att
        x = x + layer.mlp(layer.ln2(x))
    cache.advance(S0)
    logits = model.lm_head(model.gpt.final_norm(x[:, -1:]))
    out_tokens = []

    # ---- decode loop ------------------------------------------------------
    scale = 1.0 / math.sqrt(D)
    for step in range(max_new_tokens):
        if greedy:
            nxt = logits[:, -1].argmax(-1, keepdim=True)
        else:
            probs = torch.softmax(logits[:, -1].float(), -1)
            nxt = torch.multinomial(probs, 1)
        out_tokens.append(nxt)
        pos = S0 + step
        pos_ids = torch.full((B, 1), pos, device=dev, dtype=torch.long)
        x = model.gpt.embeddings(nxt, pos_ids)
        for li, layer in enumerate(model.gpt.layers):
            h = layer.ln1(x)
            q, k, v = layer_qkv(layer, h)
            cache.append(li, k, v, pos)
            cache_lens = cache.seq_lens + 1  # include the new token
            att = hot.paged_decode_attention(
                q.reshape(B, H, D), cache.k[li], cache.v[li],
                cache.block_table, cache_lens, scale)
            att = layer.attn.out_proj(att.reshape(B, 1, H * D))
            x = x + att
            x = x + layer.mlp(layer.ln2(x))
        cache.advance(1)
        logits = model.lm_head(model.gpt.final_norm(x))
    return torch.cat(out_tokens, dim=1)


def generate_llama(model, input_ids, max_new_tokens=16, cache_blocks=None,
                   block_size=16, greedy=True):
    """Greedy/sampled generation for LlamaForCausalLM with a paged GQA KV
    cache: rope applied at the absolute position, K/V cached at
    num_kv_heads, paged decode attention broadcasting HKV -> H
    (decode_attn.hip handles the head mapping in-kernel).

    Reference role: the PaddleNLP llama generation loop over the paged
    block_multihead path; re-derived here against models/llama.py."""
    model.eval()
    cfg = model.cfg
    B, S0 = input_ids.shape
    dev = input_ids.device
    H = cfg.num_heads
    HKV = cfg.num_kv_heads
    D = cfg.hidden_size // H
    max_seq = S0 + max_new_tokens
    nblocks = cache_blocks or (B * ((max_seq + block_size - 1) // block_size))
    cache = PagedKVCache(cfg.num_layers, nblocks, block_size, HKV, D, B, max_seq,
                         dev, dtype=model.lm_head.weight.dtype)

    def layer_qkv(layer, x, pos0):
        b, s, _ = x.shape
        a = layer.self_attn
        q = a.q_proj(x).reshape(b, s, H, D)
        k = a.k_proj(x).reshape(b, s, HKV, D)
        v = a.v_proj(x).reshape(b, s, HKV, D)
        q, k = hot.fused_rotary_position_embedding(q, k, base=cfg.rope_base,
                                                   pos_offset=pos0)
        return q, k, v

    # ---- prefill ----------------------------------------------------------
    x = model.llama.embed_tokens(input_ids)
    for li, layer in enumerate(model.llama.layers):
        h = layer.input_layernorm(x)
        q, k, v = layer_qkv(layer, h, 0)
        cache.append(li, k, v, 0)
        att, _ = hot.flash_attention(q, k, v, causal=True)
        x = x + layer.self_attn.o_proj(
            att.reshape(att.shape[0], att.shape[1], -1))
        x = x + layer.mlp(layer.post_attention_layernorm(x))
    cache.advance(S0)
    logits = model.lm_head(model.llama.norm(x[:, -1:]))
    out_tokens = []

    # ---- decode loop ------------------------------------------------------
    scale = 1.0 / math.sqrt(D)
    for step in range(max_new_tokens):
        if greedy:
            nxt = logits[:, -1].argmax(-1, keepdim=True)
        else:
            probs = torch.softmax(logits[:, -1].float(), -1)
            nxt = torch.multinomial(probs, 1)
        out_tokens.append(nxt)
        pos = S0 + step
        x = model.llama.embed_tokens(nxt)
        for li, layer in enumerate(model.llama.layers):
            h = layer.input_layernorm(x)
            q, k, v = layer_qkv(layer, h, pos)
            cache.append(li, k, v, pos)
            cache_lens = cache.seq_lens + 1
            att = hot.paged_decode_attention(
                q.reshape(B, H, D), cache.k[li], cache.v[li],
                cache.block_table, cache_lens, scale)
            x = x + layer.self_attn.o_proj(att.reshape(B, 1, H * D))
            x = x + layer.mlp(layer.post_attention_layernorm(x))
        cache.advance(1)
        logits = model.lm_head(model.llama.norm(x))
    return torch.cat(out_tokens, dim=1)
