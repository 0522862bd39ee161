"""HPT: Perceiver-style policy stem (learnable latent tokens cross-attending
into state features) feeding a dueling Q head.

Parity: reference ding/model/template/hpt.py (HPT:10, PolicyStem:68,
CrossAttention:145). Re-designed: attention runs through
F.scaled_dot_product_attention (maps to the fused attention kernel on ROCm)
instead of einops-expanded einsum chains.
"""
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.model.common.head import DuelingHead
from ding.utils import MODEL_REGISTRY, squeeze


class CrossAttention(nn.Module):
    """Latent tokens (queries) attend over context features (keys/values)."""

    def __init__(self, query_dim: int, heads: int = 8, dim_head: int = 64, dropout: float = 0.0):
        super().__init__()
        inner_dim = dim_head * heads
        self.heads = heads
        self.dim_head = dim_head
        self.to_q = nn.Linear(query_dim, inner_dim, bias=False)
        self.to_kv = nn.Linear(query_dim, inner_dim * 2, bias=False)
        self.to_out = nn.Linear(inner_dim, query_dim)
        self.dropout_p = dropout

    def forward(self, x: torch.Tensor, context: torch.Tensor, mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, N, _ = x.shape
        q = self.to_q(x).reshape(B, N, self.heads, self.dim_head).transpose(1, 2)
        k, v = self.to_kv(context).chunk(2, dim=-1)
        k = k.reshape(B, -1, self.heads, self.dim_head).transpose(1, 2)
        v = v.reshape(B, -1, self.heads, self.dim_head).transpose(1, 2)
        attn_mask = None
        if mask is not None:
            attn_mask = mask.reshape(B, 1, 1, -1)
        out = F.scaled_dot_product_attention(
            q, k, v, attn_mask=attn_mask, dropout_p=self.dropout_p if self.training else 0.0
        )
        out = out.transpose(1, 2).reshape(B, N, self.heads * self.dim_head)
        return self.to_out(out)


class PolicyStem(nn.Module):
    """Feature extractor + 16 learnable tokens cross-attending the features."""

    INIT_CONST = 0.02

    def __init__(self, feature_dim: int = 8, token_dim: int = 128, token_num: int = 16):
        super().__init__()
        self.feature_extractor = nn.Linear(feature_dim, token_dim)
        self.tokens = nn.Parameter(torch.randn(1, token_num, token_dim) * self.INIT_CONST)
        self.cross_attention = CrossAttention(token_dim, heads=8, dim_head=64, dropout=0.1)

    def compute_latent(self, x: torch.Tensor) -> torch.Tensor:
        stem_feat = self.feature_extractor(x)
        stem_feat = stem_feat.reshape(stem_feat.shape[0], -1, stem_feat.shape[-1])
        stem_tokens = self.tokens.expand(stem_feat.shape[0], -1, -1)
        return self.cross_attention(stem_tokens, stem_feat)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.compute_latent(x)

    @property
    def device(self) -> torch.device:
        return next(self.parameters()).device


@MODEL_REGISTRY.register('hpt')
class HPT(nn.Module):
    """Policy stem -> flatten 16x128 tokens -> dueling Q head."""

    def __init__(self, state_dim: int, action_dim: int, token_dim: int = 128, token_num: int = 16):
        super().__init__()
        self.policy_stem = PolicyStem(state_dim, token_dim, token_num)
        action_dim = squeeze(action_dim)
        self.head = DuelingHead(hidden_size=token_num * token_dim, output_size=action_dim)

    def forward(self, x: torch.Tensor):
        tokens = self.policy_stem.compute_latent(x)
        return self.head(tokens.flatten(1))
