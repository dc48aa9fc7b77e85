"""BERT encoder for the SQuAD-shape decentralized benchmark
(BASELINE.json config: "BERT-Large SQuAD-shape Decentralized-SGD").

Plain transformer encoder matching BERT-Large dimensions (24 layers,
hidden 1024, 16 heads, ffn 4096) with a span-prediction head (start/end
logits) — the training *shape* of SQuAD finetuning on synthetic data.
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 1024
    num_layers: int = 24
    num_heads: int = 16
    intermediate_size: int = 4096
    max_position: int = 512
    dropout: float = 0.1


class BertSelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.num_heads = cfg.num_heads
        self.head_dim = cfg.hidden_size // cfg.num_heads
        self.qkv = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size)
        self.out = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.dropout = nn.Dropout(cfg.dropout)

    def forward(self, x):
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.unbind(2)
        q = q.transpose(1, 2)  # B, h, S, d
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        # scaled_dot_product_attention lowers to the ROCm fused path on GPU
        y = torch.nn.functional.scaled_dot_product_attention(q, k, v)
        y = y.transpose(1, 2).reshape(B, S, H)
        return self.dropout(self.out(y))


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = BertSelfAttention(cfg)
        self.ln1 = nn.LayerNorm(cfg.hidden_size)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.hidden_size, cfg.intermediate_size),
            nn.GELU(),
            nn.Linear(cfg.intermediate_size, cfg.hidden_size),
            nn.Dropout(cfg.dropout),
        )
        self.ln2 = nn.LayerNorm(cfg.hidden_size)

    def forward(self, x):
        x = self.ln1(x + self.attn(x))
        x = self.ln2(x + self.mlp(x))
        return x


class BertForPretrainingShape(nn.Module):
    """BERT encoder + SQuAD span head; input is token ids, output is
    (start_logits, end_logits)."""

    def __init__(self, cfg: BertConfig = None):
        super().__init__()
        cfg = cfg or BertConfig()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.pos_emb = nn.Embedding(cfg.max_position, cfg.hidden_size)
        self.emb_ln = nn.LayerNorm(cfg.hidden_size)
        self.layers = nn.ModuleList(
            [BertLayer(cfg) for _ in range(cfg.num_layers)])
        self.qa_head = nn.Linear(cfg.hidden_size, 2)
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02 / math.sqrt(2))
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device).unsqueeze(0)
        x = self.emb_ln(self.tok_emb(input_ids) + self.pos_emb(pos))
        for layer in self.layers:
            x = layer(x)
        logits = self.qa_head(x)  # B, S, 2
        return logits[..., 0], logits[..., 1]


def bert_large():
    return BertForPretrainingShape(BertConfig())


def bert_tiny():
    """4-layer miniature for CPU tests."""
    return BertForPretrainingShape(BertConfig(
        vocab_size=1024, hidden_size=128, num_layers=4, num_heads=4,
        intermediate_size=256, max_position=128))
