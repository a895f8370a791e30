"""Word-level LM Transformer (parity with reference Net/Transformer.py).

Embedding(x)*sqrt(d) -> sinusoidal positional encoding + dropout ->
N post-norm encoder layers (causal MHA, FFN w/ ReLU) -> linear decoder ->
log_softmax.  Reference config: ntokens=33278, d=200, nhead=2, ffn=200,
layers=2, dropout=0.2 (dbs.py:337-343) -> 13,828,478 params.

We implement the encoder layer ourselves (instead of nn.TransformerEncoder
at Net/Transformer.py:63-64) so attention/LayerNorm/linear run through
ops.functional, where the gfx950 kernels are wired in.  Layer math matches
torch's classic post-norm TransformerEncoderLayer.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as FD
from ..ops.layers import LayerNorm, Linear


class PositionalEncoding(nn.Module):
    def __init__(self, d_model, dropout=0.1, max_len=5000):
        super().__init__()
        self.dropout = nn.Dropout(dropout)
        pos = torch.arange(max_len, dtype=torch.float).unsqueeze(1)
        freq = torch.exp(torch.arange(0, d_model, 2, dtype=torch.float)
                         * (-math.log(10000.0) / d_model))
        pe = torch.zeros(max_len, 1, d_model)
        pe[:, 0, 0::2] = torch.sin(pos * freq)
        pe[:, 0, 1::2] = torch.cos(pos * freq)
        self.register_buffer("pe", pe)

    def forward(self, x):  # [S, B, E]
        return self.dropout(x + self.pe[: x.size(0)])


class _EncoderLayer(nn.Module):
    """Post-norm encoder layer: causal MHA + ReLU FFN (torch-1.x layout)."""

    def __init__(self, d_model, nhead, dim_ff, dropout):
        super().__init__()
        self.nhead = nhead
        self.dropout_p = dropout
        self.qkv = Linear(d_model, 3 * d_model)
        self.out_proj = Linear(d_model, d_model)
        self.ffn1 = Linear(d_model, dim_ff)
        self.ffn2 = Linear(dim_ff, d_model)
        self.norm1 = LayerNorm(d_model)
        self.norm2 = LayerNorm(d_model)

    def forward(self, x):  # [S, B, E]
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        attn = FD.causal_attention(q, k, v, self.nhead,
                                   self.dropout_p, self.training)
        attn = self.out_proj(attn)
        x = self.norm1(x + FD.dropout(attn, self.dropout_p, self.training))
        ff = self.ffn2(FD.dropout(F.relu(self.ffn1(x)),
                                  self.dropout_p, self.training))
        return self.norm2(x + FD.dropout(ff, self.dropout_p, self.training))


class TransformerModel(nn.Module):
    def __init__(self, ntoken, d_model, nhead, dim_ff, nlayers, dropout=0.5):
        super().__init__()
        self.d_model = d_model
        self.embed = nn.Embedding(ntoken, d_model)
        self.pos = PositionalEncoding(d_model, dropout)
        self.layers = nn.ModuleList(
            _EncoderLayer(d_model, nhead, dim_ff, dropout) for _ in range(nlayers)
        )
        self.decoder = Linear(d_model, ntoken)
        # reference init (Net/Transformer.py:76-80)
        nn.init.uniform_(self.embed.weight, -0.1, 0.1)
        nn.init.uniform_(self.decoder.weight, -0.1, 0.1)
        nn.init.zeros_(self.decoder.bias)

    def forward_features(self, src):  # [S, B] int64 -> [S, B, d]
        """Encoder output BEFORE the decoder — the input of the fused
        LM loss head (ops.functional.lm_loss)."""
        x = FD.embedding_scaled(src, self.embed.weight,
                                math.sqrt(self.d_model))
        x = self.pos(x)
        if x.is_cuda and torch.is_autocast_enabled("cuda"):
            # keep the whole encoder in bf16 so residual adds don't
            # promote back to fp32 (embedding output is fp32)
            x = x.to(torch.bfloat16)
        for layer in self.layers:
            x = layer(x)
        return x

    def forward(self, src):  # [S, B] int64
        return FD.log_softmax(self.decoder(self.forward_features(src)),
                              dim=-1)
