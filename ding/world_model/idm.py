"""Inverse dynamics model: predict the action that produced an observed
transition (s_t, s_{t+1}) -> a_t. Used by BCO-style imitation (infer the
demonstrator's actions from state-only demonstrations) and by
inverse-dynamics auxiliary objectives.

Parity: reference ding/world_model/idm.py (InverseDynamicsModel).
"""
from typing import Dict, Optional, Union

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.model.common import ConvEncoder, DiscreteHead, FCEncoder, RegressionHead, ReparameterizationHead
from ding.utils import SequenceType, squeeze


class InverseDynamicsModel(nn.Module):
    """Encoder over the concatenated (s_t, s_{t+1}) pair plus an action head.

    ``action_space`` selects the head:
      - ``'discrete'``   -> logits over actions (argmax in ``predict_action``)
      - ``'regression'`` -> direct continuous action prediction
      - ``'reparameterization'`` -> (mu, sigma) Gaussian with tanh squash
    Vector observations are concatenated on the feature dim (in_dim = 2*obs),
    image observations on the channel dim (in_ch = 2*C).
    """

    def __init__(
        self,
        obs_shape: Union[int, SequenceType],
        action_shape: Union[int, SequenceType],
        encoder_hidden_size_list: SequenceType = [60, 80, 100, 40],
        action_space: str = "regression",
        activation: Optional[nn.Module] = None,
        norm_type: Optional[str] = None,
    ) -> None:
        super().__init__()
        if activation is None:
            activation = nn.LeakyReLU()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if isinstance(obs_shape, int) or len(obs_shape) == 1:
            in_dim = (obs_shape if isinstance(obs_shape, int) else obs_shape[0]) * 2
            self.encoder = FCEncoder(in_dim, encoder_hidden_size_list, activation=activation, norm_type=norm_type)
        elif len(obs_shape) == 3:
            pair_shape = (obs_shape[0] * 2, *obs_shape[1:])
            self.encoder = ConvEncoder(pair_shape, encoder_hidden_size_list, activation=activation, norm_type=norm_type)
        else:
            raise RuntimeError(f"unsupported obs_shape for InverseDynamicsModel: {obs_shape}")
        feat = encoder_hidden_size_list[-1]
        assert action_space in ('discrete', 'regression', 'reparameterization'), action_space
        self.action_space = action_space
        if action_space == 'discrete':
            self.header = DiscreteHead(feat, action_shape, activation=activation, norm_type=norm_type)
        elif action_space == 'regression':
            self.header = RegressionHead(feat, action_shape, final_tanh=False, activation=activation, norm_type=norm_type)
        else:
            self.header = ReparameterizationHead(
                feat, action_shape, sigma_type='conditioned', activation=activation, norm_type=norm_type
            )

    def forward(self, x: torch.Tensor) -> Dict:
        feat = self.encoder(x)
        out = self.header(feat)
        if self.action_space == 'regression':
            return {'action': out['pred']}
        if self.action_space == 'reparameterization':
            mu, sigma = out['mu'], out['sigma']
            dist = torch.distributions.Independent(torch.distributions.Normal(mu, sigma), 1)
            return {'logit': [mu, sigma], 'action': torch.tanh(dist.rsample())}
        return out  # discrete: {'logit': ...}

    def predict_action(self, x: torch.Tensor) -> Dict:
        if self.action_space == 'discrete':
            with torch.no_grad():
                return {'action': self.forward(x)['logit'].argmax(dim=-1)}
        return self.forward(x)

    def train(self, training_set: dict = None, n_epoch: int = None, learning_rate: float = None,
              weight_decay: float = 0.0):
        """With no arguments, behaves as ``nn.Module.train()``. With a
        training set, fits the model for ``n_epoch`` full-batch steps and
        returns the mean loss (cross-entropy for discrete, L1 otherwise)."""
        if training_set is None:
            return super().train(True if n_epoch is None else n_epoch)
        opt = torch.optim.AdamW(self.parameters(), lr=learning_rate, weight_decay=weight_decay)
        losses = []
        for _ in range(n_epoch):
            obs, act = training_set['obs'], training_set['action']
            if self.action_space == 'discrete':
                loss = F.cross_entropy(self.forward(obs)['logit'], act.long())
            else:
                loss = F.l1_loss(self.forward(obs)['action'], act)
            opt.zero_grad()
            loss.backward()
            opt.step()
            losses.append(loss.item())
        return float(np.mean(losses))
