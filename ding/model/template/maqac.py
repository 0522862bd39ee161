"""Multi-agent Q actor-critic models (for MASAC / multi-agent DDPG-family).

Parity: reference ding/model/template/maqac.py (DiscreteMAQAC:12,
ContinuousMAQAC:225).
"""
from typing import Dict, Optional, Union

import torch
import torch.nn as nn

from ding.model.common.head import DiscreteHead, RegressionHead, ReparameterizationHead
from ding.utils import MODEL_REGISTRY, squeeze


@MODEL_REGISTRY.register('discrete_maqac')
class DiscreteMAQAC(nn.Module):
    """Per-agent discrete actor over agent_state + (twin) Q critic over
    global_state; obs dict carries {agent_state, global_state, action_mask}."""

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        agent_obs_shape: int,
        global_obs_shape: int,
        action_shape: int,
        twin_critic: bool = False,
        actor_head_hidden_size: int = 64,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: int = 64,
        critic_head_layer_num: int = 1,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        **kwargs,
    ):
        super().__init__()
        agent_obs_shape = squeeze(agent_obs_shape)
        action_shape = squeeze(action_shape)
        act = nn.ReLU()
        self.actor = nn.Sequential(
            nn.Linear(agent_obs_shape, actor_head_hidden_size), act,
            DiscreteHead(actor_head_hidden_size, action_shape, actor_head_layer_num, norm_type=norm_type),
        )
        self.twin_critic = twin_critic
        def make_critic():
            return nn.Sequential(
                nn.Linear(squeeze(global_obs_shape), critic_head_hidden_size), nn.ReLU(),
                DiscreteHead(critic_head_hidden_size, action_shape, critic_head_layer_num, norm_type=norm_type),
            )
        self.critic = nn.ModuleList([make_critic() for _ in range(2)]) if twin_critic else make_critic()

    def forward(self, inputs: Union[torch.Tensor, Dict], mode: str) -> Dict:
        assert mode in self.mode, mode
        return getattr(self, mode)(inputs)

    @staticmethod
    def _obs(inputs: Dict) -> Dict:
        # accept both {'obs': {...}} (learn batches) and the bare obs dict
        # (collect/eval forward)
        return inputs['obs'] if 'obs' in inputs else inputs

    def compute_actor(self, inputs: Dict) -> Dict:
        obs = self._obs(inputs)
        x = self.actor(obs['agent_state'])
        return {'logit': x['logit'], 'action_mask': obs.get('action_mask')}

    def compute_critic(self, inputs: Dict) -> Dict:
        obs = self._obs(inputs)
        gs = obs['global_state']
        if 'agent_state' in obs and gs.dim() == obs['agent_state'].dim() - 1:
            # shared global state: broadcast across the agent dim so the
            # critic emits per-agent Q ([B, A, N])
            A = obs['agent_state'].shape[-2]
            gs = gs.unsqueeze(-2).expand(*gs.shape[:-1], A, gs.shape[-1])
        if self.twin_critic:
            return {'q_value': [m(gs)['logit'] for m in self.critic]}
        return {'q_value': self.critic(gs)['logit']}


@MODEL_REGISTRY.register('continuous_maqac')
class ContinuousMAQAC(nn.Module):
    """Per-agent continuous actor (regression or reparameterization) +
    (twin) Q critic over [global_state, action]."""

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        agent_obs_shape: int,
        global_obs_shape: int,
        action_shape: int,
        action_space: str,
        twin_critic: bool = False,
        actor_head_hidden_size: int = 64,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: int = 64,
        critic_head_layer_num: int = 1,
        norm_type: Optional[str] = None,
        **kwargs,
    ):
        super().__init__()
        obs_shape = squeeze(agent_obs_shape)
        global_obs_shape = squeeze(global_obs_shape)
        action_shape = squeeze(action_shape)
        self.action_shape = action_shape
        self.action_space = action_space
        assert action_space in ('regression', 'reparameterization'), action_space
        if action_space == 'regression':  # MADDPG / MATD3
            self.actor = nn.Sequential(
                nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
                RegressionHead(
                    actor_head_hidden_size, action_shape, actor_head_layer_num, final_tanh=True, norm_type=norm_type
                ),
            )
        else:  # MASAC
            self.actor = nn.Sequential(
                nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
                ReparameterizationHead(
                    actor_head_hidden_size, action_shape, actor_head_layer_num, sigma_type='conditioned',
                    norm_type=norm_type
                ),
            )
        self.twin_critic = twin_critic
        cin = global_obs_shape + action_shape
        def make_critic():
            return nn.Sequential(
                nn.Linear(cin, critic_head_hidden_size), nn.ReLU(),
                RegressionHead(critic_head_hidden_size, 1, critic_head_layer_num, final_tanh=False,
                               norm_type=norm_type),
            )
        self.critic = nn.ModuleList([make_critic() for _ in range(2)]) if twin_critic else make_critic()

    def forward(self, inputs: Union[torch.Tensor, Dict], mode: str) -> Dict:
        assert mode in self.mode, mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, inputs: Dict) -> Dict:
        obs = inputs['agent_state'] if 'agent_state' in inputs else inputs['obs']['agent_state']
        x = self.actor(obs)
        if self.action_space == 'regression':
            return {'action': x['pred']}
        return {'logit': [x['mu'], x['sigma']]}

    def compute_critic(self, inputs: Dict) -> Dict:
        obs, action = inputs['obs']['global_state'], inputs['action']
        if action.dim() == obs.dim() - 1:
            action = action.unsqueeze(-1)
        x = torch.cat([obs, action], dim=-1)
        if self.twin_critic:
            return {'q_value': [m(x)['pred'] for m in self.critic]}
        return {'q_value': self.critic(x)['pred']}
