"""BERT on the fused incubate layers (BASELINE config 2: BERT-base bf16,
fused_attention/fused_feedforward HIP path).

Reference blueprint: the fused layer API signatures in SURVEY.md A.7
(FusedMultiHeadAttention/FusedFeedForward) composed into the standard
BERT encoder.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from .. import nn
from ..incubate.nn import FusedFeedForward, FusedMultiHeadAttention
from ..nn.initializer import Normal, _apply_initializer


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    intermediate_size: int = 3072
    max_position: int = 512
    type_vocab_size: int = 2
    hidden_dropout: float = 0.0
    attn_dropout: float = 0.0
    initializer_range: float = 0.02


PRESETS = {
    "bert-base": BertConfig(),
    "bert-large": BertConfig(hidden_size=1024, num_layers=24, num_heads=16,
                             intermediate_size=4096),
    "bert-tiny": BertConfig(vocab_size=1024, hidden_size=128, num_layers=2,
                            num_heads=2, intermediate_size=256, max_position=128),
}


class BertEmbeddings(nn.Layer):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        init = Normal(0.0, cfg.initializer_range)
        self.word_embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.position_embeddings = nn.Embedding(cfg.max_position, cfg.hidden_size)
        self.token_type_embeddings = nn.Embedding(cfg.type_vocab_size, cfg.hidden_size)
        for e in (self.word_embeddings, self.position_embeddings, self.token_type_embeddings):
            _apply_initializer(init, e.weight)
        self.layer_norm = nn.LayerNorm(cfg.hidden_size)
        self.dropout = nn.Dropout(cfg.hidden_dropout)

    def forward(self, input_ids, token_type_ids=None):
        s = input_ids.shape[1]
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0).expand_as(input_ids)
        x = self.word_embeddings(input_ids) + self.position_embeddings(pos)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        x = x + self.token_type_embeddings(token_type_ids)
        return self.dropout(self.layer_norm(x))


class BertFusedLayer(nn.Layer):
    """One encoder layer on the fused HIP path (post-LN like BERT)."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = FusedMultiHeadAttention(cfg.hidden_size, cfg.num_heads,
                                            dropout_rate=cfg.hidden_dropout,
                                            attn_dropout_rate=cfg.attn_dropout,
                                            normalize_before=False)
        self.ffn = FusedFeedForward(cfg.hidden_size, cfg.intermediate_size,
                                    dropout_rate=cfg.hidden_dropout,
                                    activation="gelu", normalize_before=False)

    def forward(self, x, attn_mask=None):
        return self.ffn(self.attn(x, attn_mask))


class BertModel(nn.Layer):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.embeddings = BertEmbeddings(cfg)
        self.encoder = nn.LayerList([BertFusedLayer(cfg) for _ in range(cfg.num_layers)])
        self.pooler = nn.Linear(cfg.hidden_size, cfg.hidden_size)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        x = self.embeddings(input_ids, token_type_ids)
        mask = None
        if attention_mask is not None:
            # [b, s] 1/0 -> additive [b, 1, 1, s]
            mask = (1.0 - attention_mask.to(device=x.device, dtype=x.dtype)) * -1e4
            mask = mask.view(mask.shape[0], 1, 1, mask.shape[1])
        for layer in self.encoder:
            x = layer(x, mask)
        pooled = torch.tanh(self.pooler(x[:, 0]))
        return x, pooled

    def sharding_units(self):
        return [self.embeddings, *self.encoder, self.pooler]


class BertForPretraining(nn.Layer):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.bert = BertModel(cfg)
        self.mlm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size)
        self.nsp_head = nn.Linear(cfg.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        seq, pooled = self.bert(input_ids, token_type_ids, attention_mask)
        return self.mlm_head(seq), self.nsp_head(pooled)


def build_bert(preset="bert-tiny", **overrides):
    import dataclasses
    cfg = dataclasses.replace(PRESETS[preset], **overrides)
    return BertForPretraining(cfg)
