"""GTrXL: Gated Transformer-XL (Parisotto et al. 2019) with segment memory.

Parity: reference ding/torch_utils/network/gtrxl.py (PositionalEmbedding,
GRUGatingUnit, AttentionXL, GatedTransformerXLLayer, GTrXL).
"""
import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


class PositionalEmbedding(nn.Module):
    """Sinusoidal relative position embedding over positions [L-1 ... 0]."""

    def __init__(self, embedding_dim: int):
        super().__init__()
        self.embedding_dim = embedding_dim
        inv_freq = 1.0 / (10000 ** (torch.arange(0.0, embedding_dim, 2.0) / embedding_dim))
        self.register_buffer('inv_freq', inv_freq)

    def forward(self, pos_seq: torch.Tensor) -> torch.Tensor:
        sinusoid = torch.einsum('i,j->ij', pos_seq, self.inv_freq)
        pos_emb = torch.cat([sinusoid.sin(), sinusoid.cos()], dim=-1)
        return pos_emb.unsqueeze(0)  # [1, L, C]


class GRUGatingUnit(nn.Module):
    """GRU-style gate g(x, y) replacing residual addition."""

    def __init__(self, input_dim: int, bg: float = 2.0):
        super().__init__()
        self.Wr = nn.Linear(input_dim, input_dim, bias=False)
        self.Ur = nn.Linear(input_dim, input_dim, bias=False)
        self.Wz = nn.Linear(input_dim, input_dim, bias=False)
        self.Uz = nn.Linear(input_dim, input_dim, bias=False)
        self.Wg = nn.Linear(input_dim, input_dim, bias=False)
        self.Ug = nn.Linear(input_dim, input_dim, bias=False)
        self.bg = nn.Parameter(torch.full([input_dim], bg))

    def forward(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        r = torch.sigmoid(self.Wr(y) + self.Ur(x))
        z = torch.sigmoid(self.Wz(y) + self.Uz(x) - self.bg)
        h = torch.tanh(self.Wg(y) + self.Ug(r * x))
        return (1 - z) * x + z * h


class Memory:
    """Rolling segment memory: [layer_num+1, memory_len, B, C]."""

    def __init__(self, memory_len: int = 20, batch_size: int = 64, embedding_dim: int = 256, layer_num: int = 3,
                 memory: Optional[torch.Tensor] = None):
        self.memory_len = memory_len
        self.batch_size = batch_size
        self.embedding_dim = embedding_dim
        self.layer_num = layer_num
        self.memory = memory

    def init(self, memory: Optional[torch.Tensor] = None):
        self.memory = memory

    def update(self, hidden_state: list):
        """hidden_state: list over layers of [cur_seq, B, C]; returns new mem."""
        if self.memory is None or hidden_state is None:
            return None
        sequence_len = hidden_state[0].shape[0]
        with torch.no_grad():
            new_memory = []
            end = self.memory_len + sequence_len
            beg = max(0, end - self.memory_len)
            for i in range(self.layer_num + 1):
                m = self.memory[i]
                h = hidden_state[i]
                cat = torch.cat([m, h], dim=0)
                new_memory.append(cat[beg:end].detach())
            new_memory = torch.stack(new_memory, dim=0)
        self.memory = new_memory
        return new_memory

    def get(self):
        return self.memory


class AttentionXL(nn.Module):
    """Relative multi-head attention (Transformer-XL form)."""

    def __init__(self, input_dim: int, head_dim: int, head_num: int, dropout: nn.Module):
        super().__init__()
        self.head_num = head_num
        self.head_dim = head_dim
        self.dropout = dropout
        self.attention_kv = nn.Linear(input_dim, head_dim * head_num * 2)
        self.attention_q = nn.Linear(input_dim, head_dim * head_num)
        self.project = nn.Linear(head_dim * head_num, input_dim)
        self.project_pos = nn.Linear(input_dim, head_dim * head_num)
        self.scale = 1 / (head_dim ** 0.5)

    def _rel_shift(self, x: torch.Tensor, zero_upper: bool = False) -> torch.Tensor:
        """Shift rel-position logits into place (TXL trick)."""
        x_padded = F.pad(x, [1, 0])  # pad last dim on the left
        x_padded = x_padded.view(*x.shape[:2], x.shape[3] + 1, x.shape[2])
        x = x_padded[:, :, 1:].view_as(x)
        if zero_upper:
            ones = torch.ones((x.shape[2], x.shape[3]), device=x.device).tril(x.shape[3] - x.shape[2])
            x = x * ones[None, None]
        return x

    def forward(self, inputs: torch.Tensor, pos_embedding: torch.Tensor, full_input: torch.Tensor,
                u: torch.Tensor, v: torch.Tensor, mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        bs, cur_seq = inputs.shape[1], inputs.shape[0]
        full_seq = full_input.shape[0]
        prev_seq = full_seq - cur_seq
        kv = self.attention_kv(full_input)
        key, value = torch.chunk(kv, 2, dim=-1)  # [full, B, H*D]
        query = self.attention_q(inputs)  # [cur, B, H*D]
        r = self.project_pos(pos_embedding)  # [1, full, H*D]

        key = key.view(full_seq, bs, self.head_num, self.head_dim)
        query = query.view(cur_seq, bs, self.head_num, self.head_dim)
        value = value.view(full_seq, bs, self.head_num, self.head_dim)
        r = r.view(full_seq, self.head_num, self.head_dim)

        q_u = query + u  # content bias
        content_attn = torch.einsum('ibhd,jbhd->bhij', q_u, key)  # [B,H,cur,full]
        q_v = query + v
        position_attn = torch.einsum('ibhd,jhd->bhij', q_v, r)
        position_attn = self._rel_shift(position_attn)
        attn = (content_attn + position_attn) * self.scale
        if mask is not None and mask.any().item():
            attn = attn.masked_fill(mask[None], value=torch.finfo(attn.dtype).min)
        attn = F.softmax(attn, dim=-1)
        attn = self.dropout(attn)
        out = torch.einsum('bhij,jbhd->ibhd', attn, value)
        out = out.contiguous().view(cur_seq, bs, -1)
        return self.project(out)


class GatedTransformerXLLayer(nn.Module):

    def __init__(self, input_dim: int, head_dim: int, hidden_dim: int, head_num: int, mlp_num: int,
                 dropout: nn.Module, activation: nn.Module, gru_gating: bool = True, gru_bias: float = 2.0):
        super().__init__()
        self.dropout = dropout
        self.gating = gru_gating
        if gru_gating:
            self.gate1 = GRUGatingUnit(input_dim, gru_bias)
            self.gate2 = GRUGatingUnit(input_dim, gru_bias)
        self.attention = AttentionXL(input_dim, head_dim, head_num, dropout)
        layers = []
        dims = [input_dim] + [hidden_dim] * (mlp_num - 1) + [input_dim]
        for i in range(mlp_num):
            layers.append(nn.Linear(dims[i], dims[i + 1]))
            if i < mlp_num - 1:
                layers.append(activation)
        layers.append(self.dropout)
        self.mlp = nn.Sequential(*layers)
        self.layernorm1 = nn.LayerNorm(input_dim)
        self.layernorm2 = nn.LayerNorm(input_dim)
        self.activation = activation

    def forward(self, inputs, pos_embedding, u, v, memory, mask=None):
        full_input = torch.cat([memory, inputs], dim=0)
        x1 = self.layernorm1(full_input)
        a1 = self.dropout(self.attention(self.layernorm1(inputs), pos_embedding, x1, u, v, mask))
        a1 = self.activation(a1)
        o1 = self.gate1(inputs, a1) if self.gating else inputs + a1
        x2 = self.layernorm2(o1)
        m2 = self.dropout(self.mlp(x2))
        o2 = self.gate2(o1, m2) if self.gating else o1 + m2
        return o2


class GTrXL(nn.Module):
    """Gated TXL over [cur_seq, B, input_dim] (auto-transposes [B,T,C] in)."""

    def __init__(
        self,
        input_dim: int,
        head_dim: int = 128,
        embedding_dim: int = 256,
        head_num: int = 2,
        mlp_num: int = 2,
        layer_num: int = 3,
        memory_len: int = 64,
        dropout_ratio: float = 0.0,
        activation: nn.Module = None,
        gru_gating: bool = True,
        gru_bias: float = 2.0,
        use_embedding_layer: bool = True,
    ):
        super().__init__()
        assert embedding_dim % 2 == 0
        self.head_num = head_num
        self.head_dim = head_dim
        self.layer_num = layer_num
        self.embedding_dim = embedding_dim
        if activation is None:
            activation = nn.ReLU()
        self.activation = activation
        self.use_embedding_layer = use_embedding_layer
        if use_embedding_layer:
            self.embedding = nn.Sequential(nn.Linear(input_dim, embedding_dim), activation)
        self.pos_embedding = PositionalEmbedding(embedding_dim)
        self.memory = None
        self.memory_len = memory_len
        self.dropout = nn.Dropout(dropout_ratio)
        self.layers = nn.ModuleList([
            GatedTransformerXLLayer(
                embedding_dim, head_dim, embedding_dim, head_num, mlp_num, self.dropout, activation, gru_gating,
                gru_bias
            ) for _ in range(layer_num)
        ])
        self.u = nn.Parameter(torch.zeros(head_num, head_dim))
        self.v = nn.Parameter(torch.zeros(head_num, head_dim))
        self.att_mask = {}
        self.pos_embedding_dict = {}

    def reset_memory(self, batch_size: Optional[int] = None, state: Optional[torch.Tensor] = None):
        self.memory = Memory(self.memory_len, batch_size or 1, self.embedding_dim, self.layer_num)
        if state is not None:
            self.memory.init(state)
        elif batch_size is not None:
            device = self.u.device
            self.memory.init(
                torch.zeros(self.layer_num + 1, self.memory_len, batch_size, self.embedding_dim, device=device)
            )

    def get_memory(self):
        return None if self.memory is None else self.memory.get()

    def forward(self, x: torch.Tensor, batch_first: bool = False, return_mem: bool = True) -> dict:
        if batch_first:
            x = torch.transpose(x, 1, 0)
        cur_seq, bs = x.shape[:2]
        if self.memory is None or self.memory.get() is None or self.memory.get().shape[2] != bs:
            self.reset_memory(bs)
        memory = self.memory.get().to(x.device, x.dtype)
        if self.use_embedding_layer:
            x = self.dropout(self.embedding(x))
        prev_seq = self.memory_len
        full_seq = cur_seq + prev_seq
        if cur_seq in self.att_mask and self.att_mask[cur_seq].device == x.device:
            attn_mask = self.att_mask[cur_seq]
        else:
            attn_mask = (
                torch.triu(torch.ones((cur_seq, full_seq), device=x.device), diagonal=1 + prev_seq).bool().unsqueeze(0)
            )
            self.att_mask[cur_seq] = attn_mask
        if cur_seq in self.pos_embedding_dict and self.pos_embedding_dict[cur_seq].device == x.device:
            pos_embedding = self.pos_embedding_dict[cur_seq]
        else:
            pos_ips = torch.arange(full_seq - 1, -1, -1.0, dtype=torch.float, device=x.device)
            pos_embedding = self.pos_embedding(pos_ips)
            self.pos_embedding_dict[cur_seq] = pos_embedding
        pos_embedding = self.dropout(pos_embedding.permute(1, 0, 2))  # [full, 1, C]

        hidden_state = [x]
        out = x
        for i, layer in enumerate(self.layers):
            out = layer(out, pos_embedding, self.u, self.v, memory[i], attn_mask)
            hidden_state.append(out.clone())
        out = self.dropout(out)
        self.memory.update(hidden_state)
        if batch_first:
            out = torch.transpose(out, 1, 0)
        if return_mem:
            return {'logit': out, 'memory': self.memory.get()}
        return {'logit': out}
