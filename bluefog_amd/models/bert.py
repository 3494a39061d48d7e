# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""BERT-base encoder with a masked-LM head — for BASELINE config 5
(seqlen-512 hierarchical neighbor_allreduce). Standard architecture,
self-contained; uses torch's fused scaled_dot_product_attention, which on
ROCm lowers to the MIOpen/CK flash-attention path."""



import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["BertConfig", "BertForMaskedLM", "bert_base"]


class BertConfig:
    def __init__(
        self,
        vocab_size=30522,
        hidden_size=768,
        num_hidden_layers=12,
        num_attention_heads=12,
        intermediate_size=3072,
        max_position_embeddings=512,
        type_vocab_size=2,
        layer_norm_eps=1e-12,
        dropout=0.1,
    ):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.layer_norm_eps = layer_norm_eps
        self.dropout = dropout


class BertEmbeddings(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.word = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.position = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.token_type = nn.Embedding(cfg.type_vocab_size, cfg.hidden_size)
        self.norm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.dropout = nn.Dropout(cfg.dropout)
        self.register_buffer(
            "pos_ids", torch.arange(cfg.max_position_embeddings).unsqueeze(0), persistent=False
        )

    def forward(self, input_ids):
        s = input_ids.shape[1]
        x = (
            self.word(input_ids)
            + self.position(self.pos_ids[:, :s])
            + self.token_type(torch.zeros_like(input_ids))
        )
        return self.dropout(self.norm(x))


class BertSelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.nh = cfg.num_attention_heads
        self.hd = cfg.hidden_size // cfg.num_attention_heads
        self.qkv = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size)
        self.out = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.dropout_p = cfg.dropout

    def forward(self, x):
        B, S, H = x.shape
        qkv = self.qkv(x).reshape(B, S, 3, self.nh, self.hd).permute(2, 0, 3, 1, 4)
        q, k, v = qkv[0], qkv[1], qkv[2]
        o = F.scaled_dot_product_attention(
            q, k, v, dropout_p=self.dropout_p if self.training else 0.0
        )
        return self.out(o.transpose(1, 2).reshape(B, S, H))


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        from bluefog_amd.ops.fused_modules import FusedAddLayerNorm

        self.attn = BertSelfAttention(cfg)
        # residual join + LayerNorm as one gfx950 kernel (eager off-GPU)
        self.norm1 = FusedAddLayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.hidden_size, cfg.intermediate_size),
            nn.GELU(),
            nn.Linear(cfg.intermediate_size, cfg.hidden_size),
        )
        self.norm2 = FusedAddLayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.dropout = nn.Dropout(cfg.dropout)

    def forward(self, x):
        x = self.norm1(self.dropout(self.attn(x)), x)
        return self.norm2(self.dropout(self.mlp(x)), x)


class BertForMaskedLM(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.embeddings = BertEmbeddings(cfg)
        self.layers = nn.ModuleList(BertLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.head.weight = self.embeddings.word.weight  # weight tying
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids, labels=None):
        x = self.embeddings(input_ids)
        for layer in self.layers:
            x = layer(x)
        logits = self.head(x)
        if labels is not None:
            return F.cross_entropy(
                logits.view(-1, self.cfg.vocab_size), labels.view(-1), ignore_index=-100
            )
        return logits


def bert_base():
    return BertForMaskedLM(BertConfig())
