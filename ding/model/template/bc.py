"""Behaviour-cloning models. Parity: reference ding/model/template/bc.py."""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ..common import ConvEncoder, FCEncoder, DiscreteHead, RegressionHead, MultiHead


def _enc(obs_shape, hidden_size_list, activation, norm_type):
    obs_shape = squeeze(obs_shape)
    if isinstance(obs_shape, int) or len(obs_shape) == 1:
        return FCEncoder(squeeze(obs_shape), hidden_size_list, activation=activation, norm_type=norm_type), \
            hidden_size_list[-1]
    if min(obs_shape[1:]) < 32:
        # small maps (e.g. the 16x16 maze): the Atari 8/4/3 stack underflows
        return ConvEncoder(
            obs_shape, hidden_size_list, activation=activation, norm_type=norm_type,
            kernel_size=[3, 3, 3], stride=[2, 2, 1]
        ), hidden_size_list[-1]
    return ConvEncoder(obs_shape, hidden_size_list, activation=activation, norm_type=norm_type), hidden_size_list[-1]


@MODEL_REGISTRY.register('bc')
class DiscreteBC(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        strides: Optional[list] = None,
    ):
        super().__init__()
        action_shape = squeeze(action_shape)
        self.encoder, out = _enc(obs_shape, encoder_hidden_size_list, activation, norm_type)
        head_hidden_size = head_hidden_size or out
        if isinstance(action_shape, int):
            self.head = DiscreteHead(head_hidden_size, action_shape, head_layer_num, activation=activation,
                                     norm_type=norm_type)
        else:
            self.head = MultiHead(DiscreteHead, head_hidden_size, action_shape, layer_num=head_layer_num,
                                  activation=activation, norm_type=norm_type)

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))


@MODEL_REGISTRY.register('continuous_bc')
class ContinuousBC(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        action_space: str = 'regression',
        actor_head_hidden_size: int = 64,
        actor_head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        assert action_space in ('regression', 'reparameterization')
        self.action_space = action_space
        from ..common import ReparameterizationHead
        if action_space == 'regression':
            self.actor = nn.Sequential(
                nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
                RegressionHead(actor_head_hidden_size, action_shape, actor_head_layer_num, final_tanh=True,
                               activation=activation, norm_type=norm_type)
            )
        else:
            self.actor = nn.Sequential(
                nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
                ReparameterizationHead(actor_head_hidden_size, action_shape, actor_head_layer_num,
                                       sigma_type='conditioned', activation=activation, norm_type=norm_type)
            )

    def forward(self, x: torch.Tensor) -> Dict:
        out = self.actor(x)
        if self.action_space == 'regression':
            return {'action': out['pred']}
        return {'logit': {'mu': out['mu'], 'sigma': out['sigma']}, 'action': out['mu']}


@MODEL_REGISTRY.register('edac')
class EDAC(nn.Module):
    """SAC actor + N-ensemble Q critics (grouped conv1d EnsembleHead)."""

    def __init__(
        self,
        obs_shape: int,
        action_shape: int,
        ensemble_num: int = 10,
        actor_head_hidden_size: int = 256,
        critic_head_hidden_size: int = 256,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        **kwargs,
    ):
        super().__init__()
        from ..common import ReparameterizationHead, EnsembleHead
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        self.ensemble_num = ensemble_num
        self.actor = nn.Sequential(
            nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
            ReparameterizationHead(actor_head_hidden_size, action_shape, 1, sigma_type='conditioned',
                                   activation=activation, norm_type=norm_type)
        )
        self.critic = EnsembleHead(
            obs_shape + action_shape, 1, critic_head_hidden_size, 2, ensemble_num, activation=activation,
            norm_type=norm_type
        )

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in ('compute_actor', 'compute_critic')
        return getattr(self, mode)(inputs)

    def compute_actor(self, obs: torch.Tensor) -> Dict:
        out = self.actor(obs)
        return {'logit': [out['mu'], out['sigma']]}

    def compute_critic(self, inputs: Dict) -> Dict:
        obs, action = inputs['obs'], inputs['action']
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        x = torch.cat([obs, action], dim=-1)  # [B, O+A]
        B = x.shape[0]
        x = x.repeat(1, self.ensemble_num).unsqueeze(-1)  # [B, E*(O+A), 1]
        q = self.critic(x)['pred'].view(B, self.ensemble_num).permute(1, 0)  # [E, B]
        return {'q_value': q}


# reference also registers DiscreteBC under 'discrete_bc'
from ding.utils import MODEL_REGISTRY as _MR
if 'discrete_bc' not in _MR:
    _MR.register('discrete_bc')(DiscreteBC)
