"""QMIX: per-agent recurrent Q networks + monotonic mixing network.

Parity: reference ding/model/template/qmix.py ('qmix' registration: Mixer
with state-conditioned hypernetworks, agent DRQN sharing).
"""
from functools import reduce
from typing import Dict, List, Optional, Sequence, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.utils import MODEL_REGISTRY, squeeze
from ding.torch_utils import MLP
from .q_learning import DRQN


class Mixer(nn.Module):
    """Monotonic mixer: Q_tot = w2(s) . elu(w1(s) Q_agents + b1(s)) + b2(s),
    with |w| enforced by abs()."""

    def __init__(self, agent_num: int, state_dim: int, mixing_embed_dim: int = 32, hypernet_embed: int = 64):
        super().__init__()
        self.agent_num = agent_num
        self.state_dim = state_dim
        self.embed_dim = mixing_embed_dim
        self.hyper_w_1 = nn.Sequential(
            nn.Linear(state_dim, hypernet_embed), nn.ReLU(), nn.Linear(hypernet_embed, self.embed_dim * agent_num)
        )
        self.hyper_w_final = nn.Sequential(
            nn.Linear(state_dim, hypernet_embed), nn.ReLU(), nn.Linear(hypernet_embed, self.embed_dim)
        )
        self.hyper_b_1 = nn.Linear(state_dim, self.embed_dim)
        self.V = nn.Sequential(nn.Linear(state_dim, self.embed_dim), nn.ReLU(), nn.Linear(self.embed_dim, 1))

    def forward(self, agent_qs: torch.Tensor, states: torch.Tensor) -> torch.Tensor:
        """agent_qs [B, A], states [B, S] -> q_tot [B]."""
        bs = agent_qs.shape[0]
        states = states.reshape(-1, self.state_dim)
        agent_qs = agent_qs.reshape(-1, 1, self.agent_num)
        w1 = torch.abs(self.hyper_w_1(states)).view(-1, self.agent_num, self.embed_dim)
        b1 = self.hyper_b_1(states).view(-1, 1, self.embed_dim)
        hidden = F.elu(torch.bmm(agent_qs, w1) + b1)
        w_final = torch.abs(self.hyper_w_final(states)).view(-1, self.embed_dim, 1)
        v = self.V(states).view(-1, 1, 1)
        y = torch.bmm(hidden, w_final) + v
        return y.view(bs)


@MODEL_REGISTRY.register('qmix')
class QMix(nn.Module):
    """Shared agent DRQN + mixer. forward input (single_step or sequence):
    {'obs': {'agent_state', 'global_state', 'action_mask'}, 'prev_state',
    'action' (optional)} -> {'total_q', 'logit', 'next_state', 'action_mask'}.
    """

    def __init__(
        self,
        agent_num: int,
        obs_shape: int,
        global_obs_shape: Union[int, List[int]],
        action_shape: int,
        hidden_size_list: Sequence,
        mixer: bool = True,
        lstm_type: str = 'gru',
        activation=nn.ReLU(),
        dueling: bool = False,
    ):
        super().__init__()
        self._act = activation
        self.agent_num = agent_num
        self.action_shape = squeeze(action_shape)
        self._agent_model = DRQN(
            squeeze(obs_shape), self.action_shape, hidden_size_list, lstm_type=lstm_type, dueling=dueling,
            activation=activation
        )
        self.mixer_flag = mixer
        if mixer:
            self._mixer = Mixer(agent_num, squeeze(global_obs_shape))
        self._global_state_encoder = nn.Identity()

    def forward(self, data: dict, single_step: bool = True) -> dict:
        agent_state = data['obs']['agent_state']
        global_state = data['obs']['global_state']
        prev_state = data['prev_state']
        action = data.get('action', None)
        if single_step:
            agent_state = agent_state.unsqueeze(0)  # [1, B, A, obs]
            global_state = global_state.unsqueeze(0)
        T, B, A = agent_state.shape[:3]
        # fold agents into batch for the shared DRQN
        agent_state_f = agent_state.reshape(T, B * A, -1)
        # prev_state: list[B] where each entry is None or list[A] of {'h','c'}
        # -> flat list[B*A] for the agent-folded DRQN
        if prev_state is not None and isinstance(prev_state, list) and len(prev_state) == B:
            flat = []
            for env_states in prev_state:
                if env_states is None:
                    flat.extend([None] * A)
                else:
                    flat.extend(env_states)
            prev_state = flat
        out = self._agent_model({'obs': agent_state_f, 'prev_state': prev_state})
        logit = out['logit'].reshape(T, B, A, -1)
        next_state_flat = out['next_state']  # list[B*A]
        next_state = [next_state_flat[i * A:(i + 1) * A] for i in range(B)]
        if action is None:
            action_mask = data['obs'].get('action_mask', None)
            masked = logit.clone()
            if action_mask is not None:
                am = action_mask
                if single_step and am.dim() == 3:
                    am = am.unsqueeze(0)
                masked = masked.masked_fill(~am.bool(), -9999999)
            action = masked.argmax(dim=-1)
        agent_q_act = logit.gather(-1, action.unsqueeze(-1)).squeeze(-1)  # [T, B, A]
        if self.mixer_flag:
            gs = self._global_state_encoder(global_state)
            total_q = self._mixer(agent_q_act.reshape(T * B, A), gs.reshape(T * B, -1)).reshape(T, B)
        else:
            total_q = agent_q_act.sum(-1)
        if single_step:
            logit, total_q, action = logit.squeeze(0), total_q.squeeze(0), action.squeeze(0)
        return {
            'total_q': total_q,
            'logit': logit,
            'action': action,
            'next_state': next_state,
            'action_mask': data['obs'].get('action_mask', None),
        }
