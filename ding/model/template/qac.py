"""Q-Actor-Critic templates for continuous (DDPG/TD3/SAC) and discrete
(discrete SAC) control.

Parity: reference ding/model/template/qac.py ('continuous_qac',
'discrete_qac'). Modes: compute_actor / compute_critic.
"""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ..common import ConvEncoder, FCEncoder, DiscreteHead, RegressionHead, ReparameterizationHead


def _encoder_for(obs_shape, hidden_size_list, activation, norm_type):
    obs_shape = squeeze(obs_shape)
    if isinstance(obs_shape, int) or len(obs_shape) == 1:
        return FCEncoder(squeeze(obs_shape), hidden_size_list, activation=activation, norm_type=norm_type)
    if len(obs_shape) == 3:
        return ConvEncoder(obs_shape, hidden_size_list, activation=activation, norm_type=norm_type)
    raise RuntimeError(f"unsupported obs_shape: {obs_shape}")


@MODEL_REGISTRY.register('continuous_qac')
class ContinuousQAC(nn.Module):

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence, dict],
        action_space: str,
        twin_critic: bool = False,
        actor_head_hidden_size: int = 64,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: int = 64,
        critic_head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        encoder_hidden_size_list: Optional[Sequence] = None,
        share_encoder: bool = False,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        assert action_space in ('regression', 'reparameterization', 'hybrid')
        self.action_space = action_space
        self.twin_critic = twin_critic
        self.share_encoder = share_encoder

        pixel = not (isinstance(obs_shape, int) or len(obs_shape) == 1)
        if pixel:
            assert encoder_hidden_size_list is not None
            if share_encoder:
                self.encoder = _encoder_for(obs_shape, encoder_hidden_size_list, activation, norm_type)
            else:
                self.actor_encoder = _encoder_for(obs_shape, encoder_hidden_size_list, activation, norm_type)
                self.critic_encoder = _encoder_for(obs_shape, encoder_hidden_size_list, activation, norm_type)
            enc_out = encoder_hidden_size_list[-1]
        else:
            self.encoder = None
            enc_out = obs_shape

        if action_space == 'regression':  # DDPG/TD3 deterministic actor
            self.actor_head = nn.Sequential(
                nn.Linear(enc_out, actor_head_hidden_size), activation if isinstance(activation, nn.Module) else nn.ReLU(),
                RegressionHead(
                    actor_head_hidden_size, action_shape, actor_head_layer_num, final_tanh=True,
                    activation=activation, norm_type=norm_type
                )
            )
        elif action_space == 'reparameterization':  # SAC gaussian actor
            self.actor_head = nn.Sequential(
                nn.Linear(enc_out, actor_head_hidden_size), activation if isinstance(activation, nn.Module) else nn.ReLU(),
                ReparameterizationHead(
                    actor_head_hidden_size, action_shape, actor_head_layer_num, sigma_type='conditioned',
                    activation=activation, norm_type=norm_type
                )
            )
        else:  # hybrid (PADDPG)
            from ding.utils import EasyDict
            action_shape = EasyDict(action_shape)
            action_shape.action_args_shape = squeeze(action_shape.action_args_shape)
            action_shape.action_type_shape = squeeze(action_shape.action_type_shape)
            self._hybrid_shape = action_shape
            self.actor_head = nn.ModuleList([
                nn.Sequential(
                    nn.Linear(enc_out, actor_head_hidden_size), nn.ReLU(),
                    DiscreteHead(actor_head_hidden_size, action_shape.action_type_shape, actor_head_layer_num,
                                 activation=activation, norm_type=norm_type)
                ),
                nn.Sequential(
                    nn.Linear(enc_out, actor_head_hidden_size), nn.ReLU(),
                    RegressionHead(actor_head_hidden_size, action_shape.action_args_shape, actor_head_layer_num,
                                   final_tanh=True, activation=activation, norm_type=norm_type)
                ),
            ])

        if action_space == 'hybrid':
            critic_in = enc_out + self._hybrid_shape.action_type_shape + self._hybrid_shape.action_args_shape
        else:
            critic_in = enc_out + action_shape

        def one_critic():
            return nn.Sequential(
                nn.Linear(critic_in, critic_head_hidden_size), nn.ReLU(),
                RegressionHead(critic_head_hidden_size, 1, critic_head_layer_num, activation=activation,
                               norm_type=norm_type)
            )

        if twin_critic:
            self.critic_head = nn.ModuleList([one_critic(), one_critic()])
        else:
            self.critic_head = one_critic()

        # convenience groups for separate optimizers
        if pixel and not share_encoder:
            self.actor = nn.ModuleList([self.actor_encoder, self.actor_head])
            self.critic = nn.ModuleList([self.critic_encoder, self.critic_head])
        else:
            self.actor = self.actor_head
            self.critic = self.critic_head

    def _enc(self, x, role: str):
        if getattr(self, 'encoder', None) is None and not hasattr(self, 'actor_encoder'):
            return x
        if self.share_encoder:
            return self.encoder(x)
        return getattr(self, f'{role}_encoder')(x) if hasattr(self, f'{role}_encoder') else x

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, obs: torch.Tensor) -> Dict:
        x = self._enc(obs, 'actor')
        if self.action_space == 'regression':
            return {'action': self.actor_head(x)['pred']}
        if self.action_space == 'reparameterization':
            out = self.actor_head(x)
            return {'logit': [out['mu'], out['sigma']]}
        o_type = self.actor_head[0](x)
        o_args = self.actor_head[1](x)
        return {'logit': o_type['logit'], 'action_args': o_args['pred']}

    def compute_critic(self, inputs: Dict) -> Dict:
        obs, action = inputs['obs'], inputs['action']
        x = self._enc(obs, 'critic')
        if self.action_space == 'hybrid':
            from ding.torch_utils.network import one_hot
            t = one_hot(inputs['logit'].argmax(-1) if 'logit' in inputs else inputs['action']['action_type'],
                        self._hybrid_shape.action_type_shape)
            a = torch.cat([t, inputs['action']['action_args']], dim=-1)
        else:
            a = action
            if a.dim() == 1:
                a = a.unsqueeze(-1)
        xa = torch.cat([x, a], dim=-1)
        if self.twin_critic:
            return {'q_value': [m(xa)['pred'].squeeze(-1) for m in self.critic_head]}
        return {'q_value': self.critic_head(xa)['pred'].squeeze(-1)}


@MODEL_REGISTRY.register('discrete_qac')
class DiscreteQAC(nn.Module):

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        twin_critic: bool = False,
        actor_head_hidden_size: int = 64,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: int = 64,
        critic_head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        encoder_hidden_size_list: Optional[Sequence] = None,
        action_space: str = 'discrete',  # accepted for cfg parity; always categorical
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        self.twin_critic = twin_critic
        pixel = not (isinstance(obs_shape, int) or len(obs_shape) == 1)
        if pixel:
            self.actor_encoder = _encoder_for(obs_shape, encoder_hidden_size_list, activation, norm_type)
            self.critic_encoder = _encoder_for(obs_shape, encoder_hidden_size_list, activation, norm_type)
            enc_out = encoder_hidden_size_list[-1]
        else:
            self.actor_encoder = self.critic_encoder = None
            enc_out = obs_shape
        self.actor_head = nn.Sequential(
            nn.Linear(enc_out, actor_head_hidden_size), nn.ReLU(),
            DiscreteHead(actor_head_hidden_size, action_shape, actor_head_layer_num, activation=activation,
                         norm_type=norm_type)
        )

        def one_critic():
            return nn.Sequential(
                nn.Linear(enc_out, critic_head_hidden_size), nn.ReLU(),
                DiscreteHead(critic_head_hidden_size, action_shape, critic_head_layer_num, activation=activation,
                             norm_type=norm_type)
            )

        self.critic_head = nn.ModuleList([one_critic(), one_critic()]) if twin_critic else one_critic()
        self.actor = nn.ModuleList([m for m in (self.actor_encoder, self.actor_head) if m is not None])
        self.critic = nn.ModuleList([m for m in (self.critic_encoder, self.critic_head) if m is not None])

    def forward(self, inputs: torch.Tensor, mode: str) -> Dict:
        assert mode in self.mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, inputs: torch.Tensor) -> Dict:
        x = self.actor_encoder(inputs) if self.actor_encoder is not None else inputs
        return {'logit': self.actor_head(x)['logit']}

    def compute_critic(self, inputs: torch.Tensor) -> Dict:
        x = self.critic_encoder(inputs) if self.critic_encoder is not None else inputs
        if self.twin_critic:
            return {'q_value': [m(x)['logit'] for m in self.critic_head]}
        return {'q_value': self.critic_head(x)['logit']}
