"""BERT-large encoder (from scratch on torch.nn).

BASELINE.md config 4: BERT-large sequence/streaming gRPC with dynamic
batching and HIP-shm I/O. Standard architecture (hidden 1024, 24
layers, 16 heads), random-init weights, synthetic token inputs.
"""

import torch
import torch.nn as nn


class BertLayer(nn.Module):
    def __init__(self, hidden, heads, ffn):
        super().__init__()
        self.attn = nn.MultiheadAttention(hidden, heads, batch_first=True)
        self.ln1 = nn.LayerNorm(hidden)
        self.fc1 = nn.Linear(hidden, ffn)
        self.fc2 = nn.Linear(ffn, hidden)
        self.ln2 = nn.LayerNorm(hidden)
        self.act = nn.GELU()

    def forward(self, x):
        a, _ = self.attn(x, x, x, need_weights=False)
        x = self.ln1(x + a)
        x = self.ln2(x + self.fc2(self.act(self.fc1(x))))
        return x


class BertEncoder(nn.Module):
    """input_ids [b, s] int64 -> pooled [b, hidden] (first token)."""

    def __init__(self, vocab_size=30522, hidden=1024, layers=24, heads=16,
                 ffn=4096, max_pos=512):
        super().__init__()
        self.tok = nn.Embedding(vocab_size, hidden)
        self.pos = nn.Embedding(max_pos, hidden)
        self.ln = nn.LayerNorm(hidden)
        self.layers = nn.ModuleList(
            [BertLayer(hidden, heads, ffn) for _ in range(layers)]
        )
        self.pooler = nn.Linear(hidden, hidden)

    def forward(self, input_ids):
        s = input_ids.shape[1]
        pos_ids = torch.arange(s, device=input_ids.device)
        x = self.ln(self.tok(input_ids) + self.pos(pos_ids)[None])
        for layer in self.layers:
            x = layer(x)
        return torch.tanh(self.pooler(x[:, 0]))


def bert_large():
    return BertEncoder()


def bert_tiny():
    """Small config for CPU tests."""
    return BertEncoder(vocab_size=128, hidden=32, layers=2, heads=2, ffn=64,
                       max_pos=64)
