"""PDQN model: continuous-args net + discrete Q net over (state, args).

Parity: reference ding/model/template/pdqn.py ('pdqn').
"""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze, EasyDict
from ..common import FCEncoder, DiscreteHead, RegressionHead


@MODEL_REGISTRY.register('pdqn')
class PDQN(nn.Module):

    mode = ['compute_discrete', 'compute_continuous']

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: dict,
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        multi_pass: bool = False,
        action_mask=None,
    ):
        super().__init__()
        obs_shape = squeeze(obs_shape)
        action_shape = EasyDict(action_shape)
        self.action_type_shape = squeeze(action_shape.action_type_shape)
        self.action_args_shape = squeeze(action_shape.action_args_shape)
        hid = encoder_hidden_size_list[-1]
        self.cont_encoder = FCEncoder(obs_shape, encoder_hidden_size_list, activation=activation, norm_type=norm_type)
        self.cont_head = nn.Sequential(
            self.cont_encoder,
            RegressionHead(hid, self.action_args_shape, 1, final_tanh=True, activation=activation,
                           norm_type=norm_type)
        )
        self.dis_encoder = FCEncoder(
            obs_shape + self.action_args_shape, encoder_hidden_size_list, activation=activation, norm_type=norm_type
        )
        self.dis_head = nn.Sequential(
            self.dis_encoder,
            DiscreteHead(hid, self.action_type_shape, 1, activation=activation, norm_type=norm_type)
        )

    def forward(self, inputs, mode: str = None) -> Dict:
        if mode is None:
            # whole-model pass for collect/eval wrappers: args net then Q net
            args = self.compute_continuous(inputs)['action_args']
            return self.compute_discrete({'state': inputs, 'action_args': args})
        assert mode in self.mode
        return getattr(self, mode)(inputs)

    def compute_continuous(self, state: torch.Tensor) -> Dict:
        args = self.cont_head(state)['pred']
        return {'action_args': args}

    def compute_discrete(self, inputs: Dict) -> Dict:
        x = torch.cat([inputs['state'], inputs['action_args']], dim=-1)
        logit = self.dis_head(x)['logit']
        return {'logit': logit, 'action_args': inputs['action_args']}
