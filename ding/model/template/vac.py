"""(shared-encoder) Value-Actor-Critic template for PPO/A2C/PPG.

Parity: reference ding/model/template/vac.py ('vac' registration).
forward(x, mode) with mode in {'compute_actor', 'compute_critic',
'compute_actor_critic'}; outputs {'logit'(, 'value')}.
"""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ..common import (
    ConvEncoder, FCEncoder, DiscreteHead, MultiHead, RegressionHead, ReparameterizationHead, PopArtVHead,
)


@MODEL_REGISTRY.register('vac')
class VAC(nn.Module):

    mode = ['compute_actor', 'compute_critic', 'compute_actor_critic']

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence, dict],
        action_space: str = 'discrete',
        share_encoder: bool = True,
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        actor_head_hidden_size: Optional[int] = None,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: Optional[int] = None,
        critic_head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        sigma_type: str = 'independent',
        fixed_sigma_value: float = 0.3,
        bound_type: Optional[str] = None,
        encoder: Optional[nn.Module] = None,
        popart_head: bool = False,
        impala_cnn_encoder: bool = False,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        self.obs_shape, self.action_shape = obs_shape, action_shape
        if actor_head_hidden_size is None:
            actor_head_hidden_size = encoder_hidden_size_list[-1]
        if critic_head_hidden_size is None:
            critic_head_hidden_size = encoder_hidden_size_list[-1]
        self.impala_cnn_encoder = impala_cnn_encoder
        self.share_encoder = share_encoder

        def new_encoder():
            if impala_cnn_encoder:
                from ..common import IMPALAConvEncoder
                return IMPALAConvEncoder(obs_shape, channels=encoder_hidden_size_list[:-1] or (16, 32, 32),
                                         outsize=encoder_hidden_size_list[-1])
            if isinstance(obs_shape, int) or len(obs_shape) == 1:
                return FCEncoder(squeeze(obs_shape), encoder_hidden_size_list, activation=activation, norm_type=norm_type)
            if len(obs_shape) == 3:
                return ConvEncoder(obs_shape, encoder_hidden_size_list, activation=activation, norm_type=norm_type)
            raise RuntimeError(f"unsupported obs_shape: {obs_shape}")

        if encoder is not None:
            if share_encoder:
                self.encoder = encoder
            else:
                raise ValueError("custom encoder requires share_encoder=True")
        elif share_encoder:
            self.encoder = new_encoder()
        else:
            self.actor_encoder = new_encoder()
            self.critic_encoder = new_encoder()

        if popart_head:
            self.critic_head = PopArtVHead(
                critic_head_hidden_size, 1, critic_head_layer_num, activation=activation, norm_type=norm_type
            )
        else:
            self.critic_head = RegressionHead(
                critic_head_hidden_size, 1, critic_head_layer_num, activation=activation, norm_type=norm_type
            )

        self.action_space = action_space
        assert action_space in ('discrete', 'continuous', 'hybrid')
        if action_space == 'continuous':
            self.multi_head = False
            self.actor_head = ReparameterizationHead(
                actor_head_hidden_size, action_shape, actor_head_layer_num, sigma_type=sigma_type,
                fixed_sigma_value=fixed_sigma_value, activation=activation, norm_type=norm_type, bound_type=bound_type
            )
        elif action_space == 'discrete':
            multi_head = not isinstance(action_shape, int)
            self.multi_head = multi_head
            if multi_head:
                self.actor_head = MultiHead(
                    DiscreteHead, actor_head_hidden_size, action_shape, layer_num=actor_head_layer_num,
                    activation=activation, norm_type=norm_type
                )
            else:
                self.actor_head = DiscreteHead(
                    actor_head_hidden_size, action_shape, actor_head_layer_num, activation=activation,
                    norm_type=norm_type
                )
        else:  # hybrid
            action_shape.action_args_shape = squeeze(action_shape.action_args_shape)
            action_shape.action_type_shape = squeeze(action_shape.action_type_shape)
            self.actor_head = nn.ModuleList([
                DiscreteHead(
                    actor_head_hidden_size, action_shape.action_type_shape, actor_head_layer_num,
                    activation=activation, norm_type=norm_type
                ),
                ReparameterizationHead(
                    actor_head_hidden_size, action_shape.action_args_shape, actor_head_layer_num,
                    sigma_type=sigma_type, fixed_sigma_value=fixed_sigma_value, activation=activation,
                    norm_type=norm_type, bound_type=bound_type
                ),
            ])

        if share_encoder:
            self.actor = nn.ModuleList([self.encoder, self.actor_head])
            self.critic = nn.ModuleList([self.encoder, self.critic_head])
        else:
            self.actor = nn.ModuleList([self.actor_encoder, self.actor_head])
            self.critic = nn.ModuleList([self.critic_encoder, self.critic_head])

    def forward(self, x: torch.Tensor, mode: str) -> Dict:
        assert mode in self.mode, f"unknown mode: {mode}"
        return getattr(self, mode)(x)

    def _actor_logit(self, x: torch.Tensor):
        if self.action_space == 'discrete':
            return self.actor_head(x)['logit']
        if self.action_space == 'continuous':
            return self.actor_head(x)
        o_type = self.actor_head[0](x)
        o_args = self.actor_head[1](x)
        return {'action_type': o_type['logit'], 'action_args': o_args}

    def compute_actor(self, x: torch.Tensor) -> Dict:
        enc = self.encoder if self.share_encoder else self.actor_encoder
        return {'logit': self._actor_logit(enc(x))}

    def compute_critic(self, x: torch.Tensor) -> Dict:
        enc = self.encoder if self.share_encoder else self.critic_encoder
        out = self.critic_head(enc(x))
        if 'unnormalized_pred' in out:  # popart
            return {'value': out['pred'].squeeze(-1), 'unnormalized_value': out['unnormalized_pred'].squeeze(-1)}
        return {'value': out['pred'].squeeze(-1)}

    def compute_actor_critic(self, x: torch.Tensor) -> Dict:
        if self.share_encoder:
            e = self.encoder(x)
            ea = ec = e
        else:
            ea, ec = self.actor_encoder(x), self.critic_encoder(x)
        out = self.critic_head(ec)
        value = out['pred'].squeeze(-1)
        ret = {'logit': self._actor_logit(ea), 'value': value}
        if 'unnormalized_pred' in out:
            ret['unnormalized_value'] = out['unnormalized_pred'].squeeze(-1)
        return ret


@MODEL_REGISTRY.register('dreamervac')
class DREAMERVAC(nn.Module):
    """DreamerV3 actor-critic over RSSM latent features: ActionHead actor
    (unimix one-hot / trunc-normal) + two-hot-symlog critic.

    Parity: reference ding/model/template/vac.py DREAMERVAC:387.
    """
    mode = ['compute_actor', 'compute_critic', 'compute_actor_critic']

    def __init__(
        self,
        action_shape,
        dyn_stoch: int = 32,
        dyn_deter: int = 512,
        dyn_discrete: int = 32,
        actor_layers: int = 2,
        value_layers: int = 2,
        units: int = 512,
        act: str = 'SiLU',
        norm: str = 'LN',
        actor_dist: str = 'onehot',
        actor_init_std: float = 1.0,
        actor_min_std: float = 0.1,
        actor_max_std: float = 1.0,
        actor_temp: float = 0.1,
        action_unimix_ratio: float = 0.01,
        **kwargs,
    ) -> None:
        super().__init__()
        from ding.torch_utils.network.dreamer import ActionHead, DenseHead
        action_shape = squeeze(action_shape)
        self.action_shape = action_shape
        feat_size = dyn_stoch * dyn_discrete + dyn_deter if dyn_discrete else dyn_stoch + dyn_deter
        self.actor = ActionHead(
            feat_size, action_shape, actor_layers, units, act, norm, actor_dist,
            actor_init_std, actor_min_std, actor_max_std, actor_temp,
            outscale=1.0, unimix_ratio=action_unimix_ratio,
        )
        self.critic = DenseHead(
            feat_size, (255, ), value_layers, units, act, norm, dist='twohot_symlog', outscale=0.0
        )
