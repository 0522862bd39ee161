"""Distributional QAC (D4PG).

Parity: reference ding/model/template/qac_dist.py ('qac_dist').
"""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ..common import RegressionHead, DistributionHead


@MODEL_REGISTRY.register('qac_dist')
class QACDIST(nn.Module):

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        action_space: str = 'regression',
        critic_head_type: str = 'categorical',
        actor_head_hidden_size: int = 64,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: int = 64,
        critic_head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        v_min: float = -10,
        v_max: float = 10,
        n_atom: int = 51,
        twin_critic: bool = False,  # accepted for cfg parity; single critic
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        assert critic_head_type == 'categorical'
        self.actor = nn.Sequential(
            nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
            RegressionHead(actor_head_hidden_size, action_shape, actor_head_layer_num, final_tanh=True,
                           activation=activation, norm_type=norm_type)
        )
        self.critic = nn.Sequential(
            nn.Linear(obs_shape + action_shape, critic_head_hidden_size), nn.ReLU(),
            DistributionHead(
                critic_head_hidden_size, 1, critic_head_layer_num, n_atom=n_atom, v_min=v_min, v_max=v_max,
                activation=activation, norm_type=norm_type
            )
        )
        self.v_min, self.v_max, self.n_atom = v_min, v_max, n_atom

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, obs: torch.Tensor) -> Dict:
        return {'action': self.actor(obs)['pred']}

    def compute_critic(self, inputs: Dict) -> Dict:
        obs, action = inputs['obs'], inputs['action']
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        x = torch.cat([obs, action], dim=-1)
        out = self.critic(x)
        # distribution [B, 1, n_atom] -> [B, n_atom]; q_value [B, 1] -> [B]
        return {'q_value': out['logit'].squeeze(1), 'distribution': out['distribution'].squeeze(1)}
