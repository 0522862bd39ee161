"""Multi-agent VAC (MAPPO/HAPPO backbone).

Parity: reference ding/model/template/mavac.py ('mavac').
"""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ..common import DiscreteHead, RegressionHead, ReparameterizationHead
from ding.torch_utils import MLP


@MODEL_REGISTRY.register('mavac')
class MAVAC(nn.Module):
    """Actor over per-agent obs; centralized critic over global state.
    Modes: compute_actor / compute_critic / compute_actor_critic."""

    mode = ['compute_actor', 'compute_critic', 'compute_actor_critic']

    def __init__(
        self,
        agent_obs_shape: Union[int, Sequence],
        global_obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        agent_num: int,
        actor_hidden_size_list: Sequence = [256, 256, 128],
        critic_hidden_size_list: Sequence = [512, 512, 256],
        action_space: str = 'discrete',
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        sigma_type: str = 'independent',
        bound_type: Optional[str] = None,
    ):
        super().__init__()
        agent_obs_shape = squeeze(agent_obs_shape)
        global_obs_shape = squeeze(global_obs_shape)
        action_shape = squeeze(action_shape)
        self.agent_num = agent_num
        self.action_space = action_space
        self.actor_encoder = MLP(
            agent_obs_shape, actor_hidden_size_list[0], actor_hidden_size_list[-1],
            len(actor_hidden_size_list), activation='relu', norm_type=norm_type
        )
        self.critic_encoder = MLP(
            global_obs_shape, critic_hidden_size_list[0], critic_hidden_size_list[-1],
            len(critic_hidden_size_list), activation='relu', norm_type=norm_type
        )
        self.critic_head = RegressionHead(critic_hidden_size_list[-1], 1, 2, activation=activation,
                                          norm_type=norm_type)
        if action_space == 'discrete':
            self.actor_head = DiscreteHead(actor_hidden_size_list[-1], action_shape, 2, activation=activation,
                                           norm_type=norm_type)
        else:
            self.actor_head = ReparameterizationHead(
                actor_hidden_size_list[-1], action_shape, 2, sigma_type=sigma_type, activation=activation,
                norm_type=norm_type, bound_type=bound_type
            )
        self.actor = nn.ModuleList([self.actor_encoder, self.actor_head])
        self.critic = nn.ModuleList([self.critic_encoder, self.critic_head])

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, x: Dict) -> Dict:
        agent_state = x['agent_state']
        emb = self.actor_encoder(agent_state)
        if self.action_space == 'discrete':
            logit = self.actor_head(emb)['logit']
            mask = x.get('action_mask', None)
            if mask is not None:
                logit = logit.masked_fill(~mask.bool(), -9999999)
            return {'logit': logit}
        out = self.actor_head(emb)
        return {'logit': {'mu': out['mu'], 'sigma': out['sigma']}}

    def compute_critic(self, x: Dict) -> Dict:
        value = self.critic_head(self.critic_encoder(x['global_state']))['pred']
        return {'value': value.squeeze(-1)}

    def compute_actor_critic(self, x: Dict) -> Dict:
        out = self.compute_actor(x)
        out.update(self.compute_critic(x))
        return out
