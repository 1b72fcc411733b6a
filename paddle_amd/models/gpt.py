"""GPT model family (flagship bench model).

Blueprint parity: the reference's in-tree GPT
(test/deprecated/auto_parallel/auto_parallel_gpt_model.py:36-756 --
GPTEmbeddings / MultiHeadAttention / TransformerDecoderLayer / GPTModel /
GPTForPretraining / GPTPretrainingCriterion), rebuilt on the gfx950 hot
ops: fused LayerNorm, flash-attention (causal), bias+gelu, fused
softmax-cross-entropy.

Sizes (GPT-3 naming): 6.7B = hidden 4096 / 32 layers / 32 heads.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch

from .. import nn
from ..nn.initializer import Constant, Normal
from ..ops import functional as hot


@dataclass
class GPTConfig:
    vocab_size: int = 50304
    hidden_size: int = 1024
    num_layers: int = 24
    num_heads: int = 16
    max_seq_len: int = 2048
    intermediate_size: int = 0  # 0 -> 4*hidden
    hidden_dropout: float = 0.0
    attn_dropout: float = 0.0
    initializer_range: float = 0.02
    use_recompute: bool = False

    def __post_init__(self):
        if self.intermediate_size == 0:
            self.intermediate_size = 4 * self.hidden_size


PRESETS = {
    "gpt3-125m": GPTConfig(hidden_size=768, num_layers=12, num_heads=12),
    "gpt3-350m": GPTConfig(hidden_size=1024, num_layers=24, num_heads=16),
    "gpt3-1.3b": GPTConfig(hidden_size=2048, num_layers=24, num_heads=16),
    "gpt3-2.7b": GPTConfig(hidden_size=2560, num_layers=32, num_heads=20),
    "gpt3-6.7b": GPTConfig(hidden_size=4096, num_layers=32, num_heads=32),
    "gpt3-13b": GPTConfig(hidden_size=5120, num_layers=40, num_heads=40),
    "gpt3-tiny": GPTConfig(vocab_size=1024, hidden_size=128, num_layers=2,
                           num_heads=4, max_seq_len=256),
}


class GPTEmbeddings(nn.Layer):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        init = Normal(0.0, cfg.initializer_range)
        self.word_embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.position_embeddings = nn.Embedding(cfg.max_seq_len, cfg.hidden_size)
        from ..nn.initializer import _apply_initializer
        _apply_initializer(init, self.word_embeddings.weight)
        _apply_initializer(init, self.position_embeddings.weight)
        self.dropout = nn.Dropout(cfg.hidden_dropout)

    def forward(self, input_ids, position_ids=None):
        if position_ids is None:
            position_ids = torch.arange(input_ids.shape[1], device=input_ids.device)
            position_ids = position_ids.unsqueeze(0).expand_as(input_ids)
        x = self.word_embeddings(input_ids) + self.position_embeddings(position_ids)
        return self.dropout(x)


class GPTAttention(nn.Layer):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        h = cfg.hidden_size
        init = Normal(0.0, cfg.initializer_range)
        out_init = Normal(0.0, cfg.initializer_range / math.sqrt(2 * cfg.num_layers))
        self.num_heads = cfg.num_heads
        self.head_dim = h // cfg.num_heads
        self.qkv_proj = nn.Linear(h, 3 * h)
        self.out_proj = nn.Linear(h, h)
        from ..nn.initializer import _apply_initializer
        _apply_initializer(init, self.qkv_proj.weight)
        _apply_initializer(out_init, self.out_proj.weight)
        self.attn_dropout = cfg.attn_dropout

    def forward(self, x):
        b, s, h = x.shape
        qkv = self.qkv_proj(x).reshape(b, s, 3, self.num_heads, self.head_dim)
        # packed zero-copy path: strided q/k/v views + in-place dqkv
        out = hot.qkv_flash_attention(qkv, causal=True, dropout=self.attn_dropout,
                                      training=self.training)
        return self.out_proj(out)


class GPTMLP(nn.Layer):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        init = Normal(0.0, cfg.initializer_range)
        out_init = Normal(0.0, cfg.initializer_range / math.sqrt(2 * cfg.num_layers))
        self.fc1 = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias_attr=False)
        self.fc1_bias = self.create_parameter([cfg.intermediate_size], is_bias=True)
        self.fc2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size)
        from ..nn.initializer import _apply_initializer
        _apply_initializer(init, self.fc1.weight)
        _apply_initializer(out_init, self.fc2.weight)

    def forward(self, x):
        w1 = getattr(self.fc1, "weight", None)   # None once weight-only-quantized
        if w1 is not None and hot._own_linear_ok(x, w1):
            # own MFMA NT GEMM: fc1 bias+GELU in the kernel epilogue (aux
            # saved), fc2 dgrad carries dGELU in its epilogue
            return hot.fused_ffn_own(x, w1, self.fc1_bias,
                                     self.fc2.weight, self.fc2.bias)
        if w1 is not None and hot._fused_ffn_available(x):
            # fc1 bias+GELU in the hipBLASLt epilogue; backward fuses
            # dGELU + fc1 bias-grad into fc2's dgrad GEMM
            return hot.fused_ffn(x, w1, self.fc1_bias,
                                 self.fc2.weight, self.fc2.bias)
        # CPU / non-bf16 / weight-only path: fused bias+gelu kernel
        return self.fc2(hot.bias_gelu(self.fc1(x), self.fc1_bias))


class GPTDecoderLayer(nn.Layer):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.ln1 = nn.LayerNorm(cfg.hidden_size)
        self.attn = GPTAttention(cfg)
        self.ln2 = nn.LayerNorm(cfg.hidden_size)
        self.mlp = GPTMLP(cfg)
        self.dropout = cfg.hidden_dropout

    def forward(self, x):
        h = hot.dropout_add(self.attn(self.ln1(x)), x, self.dropout, self.training)
        return hot.dropout_add(self.mlp(self.ln2(h)), h, self.dropout, self.training)


class GPTModel(nn.Layer):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.cfg = cfg
        self.embeddings = GPTEmbeddings(cfg)
        self.layers = nn.LayerList([GPTDecoderLayer(cfg) for _ in range(cfg.num_layers)])
        self.final_norm = nn.LayerNorm(cfg.hidden_size)

    def forward(self, input_ids, position_ids=None):
        x = self.embeddings(input_ids, position_ids)
        use_rc = self.cfg.use_recompute and self.training
        if use_rc:
            from ..distributed.fleet.recompute import recompute
        for layer in self.layers:
            if use_rc and x.requires_grad:
                x = recompute(layer, x)
            else:
                x = layer(x)
        return self.final_norm(x)


class GPTForPretraining(nn.Layer):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.cfg = cfg
        self.gpt = GPTModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias_attr=False)
        from ..nn.initializer import Normal, _apply_initializer
        _apply_initializer(Normal(0.0, cfg.initializer_range), self.lm_head.weight)

    def forward(self, input_ids, position_ids=None):
        x = self.gpt(input_ids, position_ids)
        return self.lm_head(x)

    def sharding_units(self):
        """ZeRO-3 unit decomposition (one flat buffer per entry)."""
        return [self.gpt.embeddings, *self.gpt.layers,
                self.gpt.final_norm, self.lm_head]


class GPTPretrainingCriterion(nn.Layer):
    def __init__(self, ignore_index=-100):
        super().__init__()
        self.ignore_index = ignore_index

    def forward(self, logits, labels):
        loss = hot.softmax_cross_entropy(logits, labels, self.ignore_index,
                                         reduction="none")
        n_valid = (labels != self.ignore_index).sum().clamp(min=1)
        return loss.sum() / n_valid.to(loss.dtype)


def build_gpt(preset="gpt3-tiny", **overrides):
    import dataclasses
    cfg = dataclasses.replace(PRESETS[preset], **overrides)
    return GPTForPretraining(cfg)
