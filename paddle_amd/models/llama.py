"""Llama model family on the gfx950 hot ops (RMSNorm, rotary, swiglu,
GQA flash attention).

Covers BASELINE config 4 (Llama-2-70B, TP=4 PP=2 sharding-3 + recompute)
via the `tensor_parallel` / PipelineLayer construction helpers.
Blueprint parity: the reference ships no in-tree Llama (it lives in
PaddleNLP); the layer math follows the public architecture.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch

from .. import nn
from ..nn.initializer import Constant, Normal, _apply_initializer
from ..ops import functional as hot


@dataclass
class LlamaConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 32
    max_seq_len: int = 4096
    rope_base: float = 10000.0
    rms_eps: float = 1e-5
    initializer_range: float = 0.02
    use_recompute: bool = False
    tp_degree: int = 1


PRESETS = {
    "llama2-7b": LlamaConfig(),
    "llama2-13b": LlamaConfig(hidden_size=5120, intermediate_size=13824,
                              num_layers=40, num_heads=40, num_kv_heads=40),
    "llama2-70b": LlamaConfig(hidden_size=8192, intermediate_size=28672,
                              num_layers=80, num_heads=64, num_kv_heads=8),
    "llama-tiny": LlamaConfig(vocab_size=1024, hidden_size=128,
                              intermediate_size=256, num_layers=2,
                              num_heads=4, num_kv_heads=2, max_seq_len=256),
}


class LlamaAttention(nn.Layer):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        h = cfg.hidden_size
        self.cfg = cfg
        self.num_heads = cfg.num_heads // cfg.tp_degree
        self.num_kv_heads = cfg.num_kv_heads // max(1, cfg.tp_degree)
        self.head_dim = h // cfg.num_heads
        q_out = self.num_heads * self.head_dim
        kv_out = self.num_kv_heads * self.head_dim
        if cfg.tp_degree > 1:
            from ..distributed.fleet.mpu import ColumnParallelLinear, RowParallelLinear
            self.q_proj = ColumnParallelLinear(h, cfg.num_heads * self.head_dim,
                                               has_bias=False, gather_output=False)
            self.k_proj = ColumnParallelLinear(h, cfg.num_kv_heads * self.head_dim,
                                               has_bias=False, gather_output=False)
            self.v_proj = ColumnParallelLinear(h, cfg.num_kv_heads * self.head_dim,
                                               has_bias=False, gather_output=False)
            self.o_proj = RowParallelLinear(cfg.num_heads * self.head_dim, h,
                                            has_bias=False, input_is_parallel=True)
        else:
            self.q_proj = nn.Linear(h, q_out, bias_attr=False)
            self.k_proj = nn.Linear(h, kv_out, bias_attr=False)
            self.v_proj = nn.Linear(h, kv_out, bias_attr=False)
            self.o_proj = nn.Linear(q_out, h, bias_attr=False)
        for l in (self.q_proj, self.k_proj, self.v_proj, self.o_proj):
            _apply_initializer(Normal(0.0, cfg.initializer_range), l.weight)

    def forward(self, x):
        b, s, _ = x.shape
        q = self.q_proj(x).reshape(b, s, self.num_heads, self.head_dim)
        k = self.k_proj(x).reshape(b, s, self.num_kv_heads, self.head_dim)
        v = self.v_proj(x).reshape(b, s, self.num_kv_heads, self.head_dim)
        q, k = hot.fused_rotary_position_embedding(q, k, base=self.cfg.rope_base)
        out, _ = hot.flash_attention(q, k, v, causal=True)
        out = out.reshape(b, s, self.num_heads * self.head_dim)
        return self.o_proj(out)


class LlamaMLP(nn.Layer):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        h, i = cfg.hidden_size, cfg.intermediate_size // cfg.tp_degree * cfg.tp_degree
        self.tp = cfg.tp_degree
        if cfg.tp_degree > 1:
            from ..distributed.fleet.mpu import ColumnParallelLinear, RowParallelLinear
            # gate & up fused into one column-parallel matmul
            self.gate_up_proj = ColumnParallelLinear(h, 2 * cfg.intermediate_size,
                                                     has_bias=False, gather_output=False)
            self.down_proj = RowParallelLinear(cfg.intermediate_size, h,
                                               has_bias=False, input_is_parallel=True)
        else:
            self.gate_up_proj = nn.Linear(h, 2 * cfg.intermediate_size, bias_attr=False)
            self.down_proj = nn.Linear(cfg.intermediate_size, h, bias_attr=False)
        _apply_initializer(Normal(0.0, cfg.initializer_range), self.gate_up_proj.weight)
        _apply_initializer(Normal(0.0, cfg.initializer_range), self.down_proj.weight)

    def forward(self, x):
        gu = self.gate_up_proj(x)
        if self.tp > 1:
            # column-parallel packs [gate_shard | up_shard] per rank already
            half = gu.shape[-1] // 2
            act = hot.swiglu(gu[..., :half], gu[..., half:])
        else:
            act = hot.swiglu(gu)
        return self.down_proj(act)


class LlamaDecoderLayer(nn.Layer):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.input_layernorm = nn.RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.self_attn = LlamaAttention(cfg)
        self.post_attention_layernorm = nn.RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.mlp = LlamaMLP(cfg)

    def forward(self, x):
        h = x + self.self_attn(self.input_layernorm(x))
        return h + self.mlp(self.post_attention_layernorm(h))


class LlamaModel(nn.Layer):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        if cfg.tp_degree > 1:
            from ..distributed.fleet.mpu import VocabParallelEmbedding
            self.embed_tokens = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size)
        else:
            self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        _apply_initializer(Normal(0.0, cfg.initializer_range), self.embed_tokens.weight)
        self.layers = nn.LayerList([LlamaDecoderLayer(cfg) for _ in range(cfg.num_layers)])
        self.norm = nn.RMSNorm(cfg.hidden_size, cfg.rms_eps)

    def forward(self, input_ids):
        x = self.embed_tokens(input_ids)
        use_rc = self.cfg.use_recompute and self.training
        if use_rc:
            from ..distributed.fleet.recompute import recompute
        for layer in self.layers:
            if use_rc and x.requires_grad:
                x = recompute(layer, x)
            else:
                x = layer(x)
        return self.norm(x)


class LlamaForCausalLM(nn.Layer):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.llama = LlamaModel(cfg)
        if cfg.tp_degree > 1:
            from ..distributed.fleet.mpu import ColumnParallelLinear
            self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                                has_bias=False, gather_output=False)
        else:
            self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias_attr=False)
        _apply_initializer(Normal(0.0, cfg.initializer_range), self.lm_head.weight)

    def forward(self, input_ids):
        return self.lm_head(self.llama(input_ids))

    def sharding_units(self):
        return [self.llama.embed_tokens, *self.llama.layers, self.llama.norm,
                self.lm_head]


class LlamaPretrainingCriterion(nn.Layer):
    def __init__(self, tp_degree=1, ignore_index=-100):
        super().__init__()
        self.tp = tp_degree
        self.ignore_index = ignore_index
        if tp_degree > 1:
            from ..distributed.fleet.mpu import ParallelCrossEntropy
            self.pce = ParallelCrossEntropy(ignore_index=ignore_index)

    def forward(self, logits, labels):
        if self.tp > 1:
            loss = self.pce(logits, labels).squeeze(-1)
            n = (labels != self.ignore_index).sum().clamp(min=1)
            return loss.sum() / n.to(loss.dtype)
        loss = hot.softmax_cross_entropy(logits, labels, self.ignore_index,
                                         reduction="none")
        n = (labels != self.ignore_index).sum().clamp(min=1)
        return loss.sum() / n.to(loss.dtype)


def build_llama(preset="llama-tiny", **overrides):
    import dataclasses
    cfg = dataclasses.replace(PRESETS[preset], **overrides)
    return LlamaForCausalLM(cfg)


def build_llama_pp_descs(cfg: LlamaConfig, loss_fn=None):
    """LayerDesc list for PipelineLayer (config-4 TP+PP path)."""
    from ..distributed.fleet.pipeline import LayerDesc

    class _Embed(nn.Layer):
        def __init__(self):
            super().__init__()
            self.emb = nn.Embedding(cfg.vocab_size, cfg.hidden_size)

        def forward(self, ids):
            return self.emb(ids)

    class _Head(nn.Layer):
        def __init__(self):
            super().__init__()
            self.norm = nn.RMSNorm(cfg.hidden_size, cfg.rms_eps)
            self.head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias_attr=False)

        def forward(self, x):
            return self.head(self.norm(x))

    descs = [LayerDesc(_Embed)]
    for _ in range(cfg.num_layers):
        descs.append(LayerDesc(LlamaDecoderLayer, cfg))
    descs.append(LayerDesc(_Head))
    return descs
