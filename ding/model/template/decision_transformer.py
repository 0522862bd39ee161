"""Decision Transformer: GPT-style causal transformer over (R, s, a) tokens.

Parity: reference ding/model/template/decision_transformer.py ('dt').
"""
import math
from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.utils import MODEL_REGISTRY, squeeze


class MaskedCausalAttention(nn.Module):

    def __init__(self, h_dim: int, max_T: int, n_heads: int, drop_p: float):
        super().__init__()
        self.n_heads = n_heads
        self.max_T = max_T
        self.q_net = nn.Linear(h_dim, h_dim)
        self.k_net = nn.Linear(h_dim, h_dim)
        self.v_net = nn.Linear(h_dim, h_dim)
        self.proj_net = nn.Linear(h_dim, h_dim)
        self.att_drop = nn.Dropout(drop_p)
        self.proj_drop = nn.Dropout(drop_p)
        mask = torch.tril(torch.ones((max_T, max_T))).view(1, 1, max_T, max_T)
        self.register_buffer('mask', mask)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, T, C = x.shape
        N, D = self.n_heads, C // self.n_heads
        q = self.q_net(x).view(B, T, N, D).transpose(1, 2)
        k = self.k_net(x).view(B, T, N, D).transpose(1, 2)
        v = self.v_net(x).view(B, T, N, D).transpose(1, 2)
        weights = q @ k.transpose(2, 3) / math.sqrt(D)
        weights = weights.masked_fill(self.mask[..., :T, :T] == 0, float('-inf'))
        attention = self.att_drop(F.softmax(weights, dim=-1))
        out = (attention @ v).transpose(1, 2).reshape(B, T, C)
        return self.proj_drop(self.proj_net(out))


class Block(nn.Module):

    def __init__(self, h_dim: int, max_T: int, n_heads: int, drop_p: float):
        super().__init__()
        self.attention = MaskedCausalAttention(h_dim, max_T, n_heads, drop_p)
        self.mlp = nn.Sequential(
            nn.Linear(h_dim, 4 * h_dim), nn.GELU(), nn.Linear(4 * h_dim, h_dim), nn.Dropout(drop_p)
        )
        self.ln1 = nn.LayerNorm(h_dim)
        self.ln2 = nn.LayerNorm(h_dim)

    def forward(self, x):
        x = self.ln1(x + self.attention(x))
        x = self.ln2(x + self.mlp(x))
        return x


@MODEL_REGISTRY.register('dt')
class DecisionTransformer(nn.Module):

    def __init__(
        self,
        state_dim: int,
        act_dim: int,
        n_blocks: int = 3,
        h_dim: int = 128,
        context_len: int = 20,
        n_heads: int = 1,
        drop_p: float = 0.1,
        max_timestep: int = 4096,
        state_encoder: Optional[nn.Module] = None,
        continuous: bool = False,
    ):
        super().__init__()
        self.state_dim = state_dim
        self.act_dim = act_dim
        self.h_dim = h_dim
        self.continuous = continuous
        input_seq_len = 3 * context_len
        self.blocks = nn.ModuleList([Block(h_dim, input_seq_len, n_heads, drop_p) for _ in range(n_blocks)])
        self.embed_ln = nn.LayerNorm(h_dim)
        self.embed_timestep = nn.Embedding(max_timestep, h_dim)
        self.embed_rtg = nn.Linear(1, h_dim)
        if state_encoder is None:
            self.embed_state = nn.Linear(squeeze(state_dim), h_dim)
        else:
            self.embed_state = state_encoder
        if continuous:
            self.embed_action = nn.Linear(act_dim, h_dim)
            self.predict_action = nn.Sequential(nn.Linear(h_dim, act_dim), nn.Tanh())
        else:
            self.embed_action = nn.Embedding(act_dim, h_dim)
            self.predict_action = nn.Linear(h_dim, act_dim)
        self.predict_rtg = nn.Linear(h_dim, 1)
        self.predict_state = nn.Linear(h_dim, squeeze(state_dim) if isinstance(state_dim, (int, )) else h_dim)

    def forward(
        self, timesteps: torch.Tensor, states: torch.Tensor, actions: torch.Tensor, returns_to_go: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        B, T = states.shape[0], states.shape[1]
        time_emb = self.embed_timestep(timesteps)
        state_emb = self.embed_state(states) + time_emb
        if self.continuous:
            act_emb = self.embed_action(actions) + time_emb
        else:
            act_emb = self.embed_action(actions.long().squeeze(-1) if actions.dim() == 3 else actions.long()) + time_emb
        rtg_emb = self.embed_rtg(returns_to_go) + time_emb
        # interleave tokens (R_t, s_t, a_t)
        h = torch.stack([rtg_emb, state_emb, act_emb], dim=2).reshape(B, 3 * T, self.h_dim)
        h = self.embed_ln(h)
        for block in self.blocks:
            h = block(h)
        h = h.reshape(B, T, 3, self.h_dim).permute(0, 2, 1, 3)
        return_preds = self.predict_rtg(h[:, 2])
        state_preds = self.predict_state(h[:, 2])
        action_preds = self.predict_action(h[:, 1])
        return state_preds, action_preds, return_preds
