"""ACER model: discrete actor + per-action Q critic.

Parity: reference ding/model/template/acer.py ('acer').
"""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ..common import ConvEncoder, FCEncoder, DiscreteHead


@MODEL_REGISTRY.register('acer')
class ACER(nn.Module):

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        actor_head_hidden_size: Optional[int] = None,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: Optional[int] = None,
        critic_head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        actor_head_hidden_size = actor_head_hidden_size or encoder_hidden_size_list[-1]
        critic_head_hidden_size = critic_head_hidden_size or encoder_hidden_size_list[-1]

        def enc():
            if isinstance(obs_shape, int) or len(obs_shape) == 1:
                return FCEncoder(squeeze(obs_shape), encoder_hidden_size_list, activation=activation,
                                 norm_type=norm_type)
            return ConvEncoder(obs_shape, encoder_hidden_size_list, activation=activation, norm_type=norm_type)

        self.actor_encoder, self.critic_encoder = enc(), enc()
        self.actor_head = DiscreteHead(actor_head_hidden_size, action_shape, actor_head_layer_num,
                                       activation=activation, norm_type=norm_type)
        self.critic_head = DiscreteHead(critic_head_hidden_size, action_shape, critic_head_layer_num,
                                        activation=activation, norm_type=norm_type)
        self.actor = nn.ModuleList([self.actor_encoder, self.actor_head])
        self.critic = nn.ModuleList([self.critic_encoder, self.critic_head])

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, x: torch.Tensor) -> Dict:
        return {'logit': self.actor_head(self.actor_encoder(x))['logit']}

    def compute_critic(self, x: torch.Tensor) -> Dict:
        return {'q_value': self.critic_head(self.critic_encoder(x))['logit']}
