"""Policy-gradient model (actor only). Parity: reference
ding/model/template/pg.py ('pg')."""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ..common import ConvEncoder, FCEncoder, DiscreteHead, ReparameterizationHead, MultiHead


@MODEL_REGISTRY.register('pg')
class PG(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        action_space: str = 'discrete',
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        if isinstance(obs_shape, int) or len(obs_shape) == 1:
            self.encoder = FCEncoder(squeeze(obs_shape), encoder_hidden_size_list, activation=activation, norm_type=norm_type)
        elif len(obs_shape) == 3:
            self.encoder = ConvEncoder(obs_shape, encoder_hidden_size_list, activation=activation, norm_type=norm_type)
        else:
            raise RuntimeError(f"unsupported obs_shape: {obs_shape}")
        self.action_space = action_space
        if action_space == 'discrete':
            if isinstance(action_shape, int):
                self.head = DiscreteHead(head_hidden_size, action_shape, head_layer_num, activation=activation,
                                         norm_type=norm_type)
            else:
                self.head = MultiHead(DiscreteHead, head_hidden_size, action_shape, layer_num=head_layer_num,
                                      activation=activation, norm_type=norm_type)
        else:
            self.head = ReparameterizationHead(
                head_hidden_size, action_shape, head_layer_num, sigma_type='independent', activation=activation,
                norm_type=norm_type
            )

    def forward(self, x: torch.Tensor) -> Dict:
        x = self.encoder(x)
        out = self.head(x)
        if self.action_space == 'discrete':
            out['dist'] = torch.distributions.Categorical(logits=out['logit'])
            return out
        return {'logit': {'mu': out['mu'], 'sigma': out['sigma']}}
