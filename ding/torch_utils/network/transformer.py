"""Post-LN Transformer encoder (AlphaStar-style entity encoder).

Parity: reference ding/torch_utils/network/transformer.py (Attention,
TransformerLayer, Transformer, ScaledDotProductAttention).
"""
import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .nn_module import fc_block, build_activation


class Attention(nn.Module):
    """Multi-head attention with a single packed QKV projection."""

    def __init__(self, input_dim: int, head_dim: int, output_dim: int, head_num: int, dropout: nn.Module):
        super().__init__()
        self.head_num = head_num
        self.head_dim = head_dim
        self.dropout = dropout
        self.attention_pre = fc_block(input_dim, head_dim * head_num * 3)
        self.project = fc_block(head_dim * head_num, output_dim)

    def split(self, x: torch.Tensor, T: bool = False):
        B, N = x.shape[:2]
        x = x.view(B, N, self.head_num, self.head_dim).permute(0, 2, 1, 3).contiguous()
        if T:
            x = x.transpose(-2, -1)
        return x

    def forward(self, x: torch.Tensor, mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        assert x.dim() == 3  # [B, N, C]
        qkv = self.attention_pre(x)
        q, k, v = torch.chunk(qkv, 3, dim=2)
        q, k, v = self.split(q), self.split(k, T=True), self.split(v)
        score = torch.matmul(q, k) / math.sqrt(self.head_dim)  # [B, H, N, N]
        if mask is not None:
            score = score.masked_fill(~mask, value=torch.finfo(score.dtype).min)
        score = F.softmax(score, dim=-1)
        score = self.dropout(score)
        attn = torch.matmul(score, v)  # [B, H, N, D]
        attn = attn.permute(0, 2, 1, 3).contiguous()
        B, N = attn.shape[:2]
        return self.project(attn.view(B, N, -1))


class TransformerLayer(nn.Module):

    def __init__(self, input_dim: int, head_dim: int, hidden_dim: int, output_dim: int, head_num: int,
                 mlp_num: int, dropout: nn.Module, activation: nn.Module):
        super().__init__()
        self.attention = Attention(input_dim, head_dim, output_dim, head_num, dropout)
        self.layernorm1 = nn.LayerNorm(output_dim)
        self.dropout = dropout
        layers = []
        dims = [output_dim] + [hidden_dim] * (mlp_num - 1) + [output_dim]
        for i in range(mlp_num):
            layers.append(fc_block(dims[i], dims[i + 1], activation=activation if i < mlp_num - 1 else None))
        self.mlp = nn.Sequential(*layers)
        self.layernorm2 = nn.LayerNorm(output_dim)

    def forward(self, inputs):
        x, mask = inputs
        a = self.dropout(self.attention(x, mask))
        x = self.layernorm1(x + a)
        m = self.dropout(self.mlp(x))
        x = self.layernorm2(x + m)
        return (x, mask)


class Transformer(nn.Module):
    """Stack of post-LN transformer layers over entity sets [B, N, C]."""

    def __init__(
        self,
        input_dim: int,
        head_dim: int = 128,
        hidden_dim: int = 1024,
        output_dim: int = 256,
        head_num: int = 2,
        mlp_num: int = 2,
        layer_num: int = 3,
        dropout_ratio: float = 0.0,
        activation: str = 'relu',
    ):
        super().__init__()
        act = build_activation(activation)
        self.embedding = fc_block(input_dim, output_dim, activation=act)
        self.dropout = nn.Dropout(dropout_ratio)
        layers = [
            TransformerLayer(output_dim, head_dim, hidden_dim, output_dim, head_num, mlp_num, self.dropout, act)
            for _ in range(layer_num)
        ]
        self.main = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor, mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        if mask is not None:
            x = x * mask.unsqueeze(-1)
            mask = mask.unsqueeze(1).repeat(1, mask.shape[1], 1).unsqueeze(1)  # [B,1,N,N]
        x = self.embedding(x)
        x = self.dropout(x)
        x, _ = self.main((x, mask))
        return x


class ScaledDotProductAttention(nn.Module):

    def __init__(self, d_k: int, dropout: float = 0.0):
        super().__init__()
        self.d_k = d_k
        self.dropout = nn.Dropout(dropout)

    def forward(self, q, k, v, mask: Optional[torch.Tensor] = None):
        score = torch.matmul(q, k.transpose(-2, -1)) / math.sqrt(self.d_k)
        if mask is not None:
            score = score.masked_fill(~mask, value=torch.finfo(score.dtype).min)
        attn = self.dropout(F.softmax(score, dim=-1))
        return torch.matmul(attn, v)
