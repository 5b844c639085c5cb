"""BERT-large encoder (random init, synthetic data) for the DDP benchmark
(BERT-large, cgx_hook layerwise filter, 8-bit)."""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


class BertLayer(nn.Module):
    def __init__(self, hidden: int, heads: int, intermediate: int):
        super().__init__()
        self.heads = heads
        self.head_dim = hidden // heads
        self.qkv = nn.Linear(hidden, 3 * hidden)
        self.proj = nn.Linear(hidden, hidden)
        self.ln1 = nn.LayerNorm(hidden)
        self.fc1 = nn.Linear(hidden, intermediate)
        self.fc2 = nn.Linear(intermediate, hidden)
        self.ln2 = nn.LayerNorm(hidden)

    def forward(self, x):
        b, s, h = x.shape
        qkv = self.qkv(x).view(b, s, 3, self.heads, self.head_dim)
        q, k, v = qkv.unbind(2)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        attn = F.scaled_dot_product_attention(q, k, v)
        attn = attn.transpose(1, 2).reshape(b, s, h)
        x = self.ln1(x + self.proj(attn))
        x = self.ln2(x + self.fc2(F.gelu(self.fc1(x))))
        return x


class Bert(nn.Module):
    def __init__(self, vocab: int = 30522, hidden: int = 1024,
                 layers: int = 24, heads: int = 16, intermediate: int = 4096,
                 max_len: int = 512):
        super().__init__()
        self.tok = nn.Embedding(vocab, hidden)
        self.pos = nn.Embedding(max_len, hidden)
        self.ln = nn.LayerNorm(hidden)
        self.blocks = nn.ModuleList(
            [BertLayer(hidden, heads, intermediate) for _ in range(layers)])
        self.head = nn.Linear(hidden, vocab, bias=False)
        self.head.weight = self.tok.weight  # tied
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, ids):
        b, s = ids.shape
        pos = torch.arange(s, device=ids.device).unsqueeze(0)
        x = self.ln(self.tok(ids) + self.pos(pos))
        for blk in self.blocks:
            x = blk(x)
        return self.head(x)


def bert_large(vocab: int = 30522) -> Bert:
    return Bert(vocab=vocab, hidden=1024, layers=24, heads=16,
                intermediate=4096)


def bert_tiny(vocab: int = 1000) -> Bert:
    """Small variant for tests."""
    return Bert(vocab=vocab, hidden=64, layers=2, heads=2, intermediate=128,
                max_len=128)
