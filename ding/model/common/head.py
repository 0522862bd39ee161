"""Prediction heads. Output conventions match the reference exactly
(ding/model/common/head.py): DiscreteHead->{'logit'}, DistributionHead->
{'logit','distribution'}, QRDQNHead->{'logit','q','tau'}, QuantileHead->
{'logit','q','quantiles'}, RegressionHead->{'pred'},
ReparameterizationHead->{'mu','sigma'}, etc.
"""
import math
from typing import Dict, Optional

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.torch_utils import MLP, fc_block, noise_block, NoisyLinearLayer, PopArt
from ding.rl_utils import beta_function_map


class DiscreteHead(nn.Module):
    """Q-value / logit head for discrete action spaces."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int,
        layer_num: int = 1,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        dropout: Optional[float] = None,
        noise: bool = False,
    ):
        super().__init__()
        block = noise_block if noise else fc_block
        layers = []
        for _ in range(layer_num):
            layers.append(
                fc_block(hidden_size, hidden_size, activation=activation, norm_type=norm_type,
                         use_dropout=dropout is not None, dropout_probability=dropout or 0.5)
                if not noise else
                noise_block(hidden_size, hidden_size, activation=activation, norm_type=norm_type,
                            use_dropout=dropout is not None, dropout_probability=dropout or 0.5)
            )
        last = noise_block(hidden_size, output_size) if noise else fc_block(hidden_size, output_size)
        self.Q = nn.Sequential(*layers, last)

    def forward(self, x: torch.Tensor) -> Dict:
        return {'logit': self.Q(x)}


class DistributionHead(nn.Module):
    """C51 head: per-action categorical distribution over the value support."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int,
        layer_num: int = 1,
        n_atom: int = 51,
        v_min: float = -10,
        v_max: float = 10,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        noise: bool = False,
        eps: float = 1e-6,
    ):
        super().__init__()
        block = noise_block if noise else fc_block
        layers = [block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(layer_num)]
        layers.append(block(hidden_size, output_size * n_atom))
        self.Q = nn.Sequential(*layers)
        self.output_size = output_size
        self.n_atom = n_atom
        self.v_min = v_min
        self.v_max = v_max
        self.eps = eps

    def forward(self, x: torch.Tensor) -> Dict:
        q = self.Q(x).view(*x.shape[:-1], self.output_size, self.n_atom)
        dist = torch.softmax(q, dim=-1) + self.eps
        support = torch.linspace(self.v_min, self.v_max, self.n_atom, device=x.device)
        q_val = (dist * support).sum(-1)
        return {'logit': q_val, 'distribution': dist}


class RainbowHead(nn.Module):
    """Dueling + distributional + noisy head (Rainbow)."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int,
        layer_num: int = 1,
        n_atom: int = 51,
        v_min: float = -10,
        v_max: float = 10,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        noise: bool = True,
        eps: float = 1e-6,
    ):
        super().__init__()
        block = noise_block if noise else fc_block
        self.A = nn.Sequential(
            *[block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(layer_num)],
            block(hidden_size, output_size * n_atom)
        )
        self.Q = nn.Sequential(
            *[block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(layer_num)],
            block(hidden_size, n_atom)
        )
        self.output_size = output_size
        self.n_atom = n_atom
        self.v_min = v_min
        self.v_max = v_max
        self.eps = eps

    def forward(self, x: torch.Tensor) -> Dict:
        a = self.A(x).view(*x.shape[:-1], self.output_size, self.n_atom)
        v = self.Q(x).view(*x.shape[:-1], 1, self.n_atom)
        dist_logits = v + a - a.mean(dim=-2, keepdim=True)
        dist = torch.softmax(dist_logits, dim=-1) + self.eps
        support = torch.linspace(self.v_min, self.v_max, self.n_atom, device=x.device)
        q = (dist * support).sum(-1)
        return {'logit': q, 'distribution': dist}


class QRDQNHead(nn.Module):
    """Quantile-regression DQN head (fixed uniform fractions)."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int,
        layer_num: int = 1,
        num_quantiles: int = 32,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        self.Q = nn.Sequential(
            *[fc_block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(layer_num)],
            fc_block(hidden_size, output_size * num_quantiles)
        )
        self.num_quantiles = num_quantiles
        self.output_size = output_size

    def forward(self, x: torch.Tensor) -> Dict:
        q = self.Q(x).view(*x.shape[:-1], self.output_size, self.num_quantiles)
        logit = q.mean(-1)
        tau = torch.linspace(0, 1, self.num_quantiles + 1, device=x.device)
        tau = ((tau[:-1] + tau[1:]) / 2).view(1, -1, 1).repeat(x.shape[0], 1, 1)
        return {'logit': logit, 'q': q.permute(0, 2, 1), 'tau': tau}  # q: [B, tau, N]


class QuantileHead(nn.Module):
    """IQN head: sampled quantile fractions with cosine embeddings."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int,
        layer_num: int = 1,
        num_quantiles: int = 32,
        quantile_embedding_size: int = 128,
        beta_function_type: str = 'uniform',
        activation: str = 'relu',
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        self.Q = nn.Sequential(
            *[fc_block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(layer_num)],
            fc_block(hidden_size, output_size)
        )
        self.num_quantiles = num_quantiles
        self.quantile_embedding_size = quantile_embedding_size
        self.output_size = output_size
        self.iqn_fc = nn.Linear(quantile_embedding_size, hidden_size)
        self.beta_function = beta_function_map[beta_function_type]

    def quantile_net(self, quantiles: torch.Tensor) -> torch.Tensor:
        i = torch.arange(1, self.quantile_embedding_size + 1, device=quantiles.device).float()
        cos = torch.cos(math.pi * i.view(1, -1) * quantiles)  # [Nq*B, E]
        return F.relu(self.iqn_fc(cos))

    def forward(self, x: torch.Tensor, num_quantiles: Optional[int] = None) -> Dict:
        if num_quantiles is None:
            num_quantiles = self.num_quantiles
        B = x.shape[0]
        q_quantiles = torch.rand(num_quantiles * B, 1, device=x.device)
        logit_quantiles = self.beta_function(q_quantiles)
        quantile_emb = self.quantile_net(logit_quantiles)  # [Nq*B, H]
        x_rep = x.repeat(num_quantiles, 1)  # [Nq*B, H]
        q = self.Q(x_rep * quantile_emb).view(num_quantiles, B, self.output_size)
        logit = q.mean(0)
        return {'logit': logit, 'q': q, 'quantiles': q_quantiles}


class FQFHead(nn.Module):
    """FQF head: learned fraction proposal + quantile value net."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int,
        layer_num: int = 1,
        num_quantiles: int = 32,
        quantile_embedding_size: int = 128,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        self.Q = nn.Sequential(
            *[fc_block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(layer_num)],
            fc_block(hidden_size, output_size)
        )
        self.num_quantiles = num_quantiles
        self.quantile_embedding_size = quantile_embedding_size
        self.output_size = output_size
        self.fqf_fc = nn.Sequential(nn.Linear(hidden_size, num_quantiles), nn.LogSoftmax(dim=-1))
        self.register_buffer(
            'sigma_pi', torch.arange(1, quantile_embedding_size + 1, 1).float() * math.pi
        )
        self.quantile_fc = nn.Linear(quantile_embedding_size, hidden_size)

    def quantile_net(self, quantiles: torch.Tensor) -> torch.Tensor:
        cos = torch.cos(quantiles.unsqueeze(-1) * self.sigma_pi)  # [B, Nq, E]
        return F.relu(self.quantile_fc(cos))

    def forward(self, x: torch.Tensor, num_quantiles: Optional[int] = None) -> Dict:
        B = x.shape[0]
        log_q = self.fqf_fc(x.detach())  # fraction proposal on detached features
        q_prob = log_q.exp()
        quantiles = torch.cumsum(q_prob, dim=1)
        quantiles = torch.cat([torch.zeros(B, 1, device=x.device), quantiles], dim=1)  # [B, Nq+1]
        quantiles_hats = ((quantiles[:, 1:] + quantiles[:, :-1]) / 2).detach()  # [B, Nq]
        emb = self.quantile_net(quantiles_hats)  # [B, Nq, H]
        q = self.Q(x.unsqueeze(1) * emb)  # [B, Nq, N]
        logit = (q_prob.unsqueeze(-1).detach() * q).sum(1)
        with torch.no_grad():
            emb_i = self.quantile_net(quantiles[:, 1:-1])
            q_tau_i = self.Q(x.unsqueeze(1) * emb_i)  # [B, Nq-1, N]
        return {'logit': logit, 'q': q, 'quantiles': quantiles, 'quantiles_hats': quantiles_hats,
                'q_tau_i': q_tau_i, 'entropies': -(log_q * q_prob).sum(-1, keepdim=True)}


class DuelingHead(nn.Module):
    """Dueling architecture: Q = V + A - mean(A)."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int,
        layer_num: int = 1,
        a_layer_num: Optional[int] = None,
        v_layer_num: Optional[int] = None,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        dropout: Optional[float] = None,
        noise: bool = False,
    ):
        super().__init__()
        a_layer_num = a_layer_num or layer_num
        v_layer_num = v_layer_num or layer_num
        block = noise_block if noise else fc_block
        self.A = nn.Sequential(
            *[block(hidden_size, hidden_size, activation=activation, norm_type=norm_type,
                    use_dropout=dropout is not None, dropout_probability=dropout or 0.5) for _ in range(a_layer_num)],
            block(hidden_size, output_size)
        )
        self.V = nn.Sequential(
            *[block(hidden_size, hidden_size, activation=activation, norm_type=norm_type,
                    use_dropout=dropout is not None, dropout_probability=dropout or 0.5) for _ in range(v_layer_num)],
            block(hidden_size, 1)
        )

    def forward(self, x: torch.Tensor) -> Dict:
        a = self.A(x)
        v = self.V(x)
        return {'logit': a - a.mean(dim=-1, keepdim=True) + v}


class BranchingHead(nn.Module):
    """BDQ: shared value + per-branch advantages over discretized sub-actions."""

    def __init__(
        self,
        hidden_size: int,
        num_branches: int = 0,
        action_bins_per_branch: int = 2,
        layer_num: int = 1,
        a_layer_num: Optional[int] = None,
        v_layer_num: Optional[int] = None,
        norm_type: Optional[str] = None,
        activation: str = 'relu',
        noise: bool = False,
    ):
        super().__init__()
        a_layer_num = a_layer_num or layer_num
        v_layer_num = v_layer_num or layer_num
        self.num_branches = num_branches
        self.action_bins_per_branch = action_bins_per_branch
        block = noise_block if noise else fc_block
        self.branches = nn.ModuleList([
            nn.Sequential(
                *[block(hidden_size, hidden_size, activation=activation, norm_type=norm_type)
                  for _ in range(a_layer_num)],
                block(hidden_size, action_bins_per_branch)
            ) for _ in range(num_branches)
        ])
        self.V = nn.Sequential(
            *[block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(v_layer_num)],
            block(hidden_size, 1)
        )

    def forward(self, x: torch.Tensor) -> Dict:
        value = self.V(x).unsqueeze(1)
        advs = torch.stack([b(x) for b in self.branches], dim=1)  # [B, D, bins]
        q = value + advs - advs.mean(dim=2, keepdim=True)
        return {'logit': q}


class StochasticDuelingHead(nn.Module):
    """Continuous-action dueling (ACER continuous): V plus sampled-advantage."""

    def __init__(
        self,
        hidden_size: int,
        action_shape: int,
        layer_num: int = 1,
        a_layer_num: Optional[int] = None,
        v_layer_num: Optional[int] = None,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        noise: bool = False,
        last_tanh: bool = True,
    ):
        super().__init__()
        a_layer_num = a_layer_num or layer_num
        v_layer_num = v_layer_num or layer_num
        block = noise_block if noise else fc_block
        self.A = nn.Sequential(
            *[block(hidden_size + action_shape, hidden_size + action_shape, activation=activation,
                    norm_type=norm_type) for _ in range(a_layer_num)],
            block(hidden_size + action_shape, 1)
        )
        self.V = nn.Sequential(
            *[block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(v_layer_num)],
            block(hidden_size, 1)
        )
        self.tanh = nn.Tanh() if last_tanh else None

    def forward(
        self,
        s: torch.Tensor,
        a: torch.Tensor,
        mu: torch.Tensor,
        sigma: torch.Tensor,
        sample_size: int = 10,
    ) -> Dict:
        B = s.shape[0]
        v = self.V(s)
        adv_taken = self.A(torch.cat([s, a], dim=-1))
        dist = torch.distributions.Normal(mu, sigma)
        samples = dist.sample((sample_size, ))  # [K, B, D]
        s_rep = s.unsqueeze(0).expand(sample_size, *s.shape)
        adv_mean = self.A(torch.cat([s_rep, samples], dim=-1)).mean(0)
        q = v + adv_taken - adv_mean
        if self.tanh is not None:
            pass  # tanh applies to action pre-processing upstream in reference
        return {'q_value': q, 'v_value': v}


class RegressionHead(nn.Module):
    """Continuous regression (DDPG actor / Q(s,a) critic): {'pred'}."""

    def __init__(
        self,
        input_size: int,
        output_size: int,
        layer_num: int = 2,
        final_tanh: bool = False,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        hidden_size: Optional[int] = None,
    ):
        super().__init__()
        hidden_size = hidden_size or input_size
        self.main = MLP(input_size, hidden_size, hidden_size, layer_num, activation=activation, norm_type=norm_type)
        self.last = nn.Linear(hidden_size, output_size)
        self.final_tanh = final_tanh
        if final_tanh:
            self.tanh = nn.Tanh()

    def forward(self, x: torch.Tensor) -> Dict:
        x = self.last(self.main(x))
        if self.final_tanh:
            x = self.tanh(x)
        return {'pred': x}


class ReparameterizationHead(nn.Module):
    """Gaussian policy head: {'mu','sigma'}; sigma_type in
    {'fixed','independent','conditioned','happo'}."""

    def __init__(
        self,
        input_size: int,
        output_size: int,
        layer_num: int = 2,
        sigma_type: str = 'independent',
        fixed_sigma_value: float = 1.0,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        bound_type: Optional[str] = None,
        hidden_size: Optional[int] = None,
    ):
        super().__init__()
        hidden_size = hidden_size or input_size
        assert sigma_type in ('fixed', 'independent', 'conditioned', 'happo')
        self.sigma_type = sigma_type
        self.bound_type = bound_type
        self.main = MLP(input_size, hidden_size, hidden_size, layer_num, activation=activation, norm_type=norm_type)
        self.mu = nn.Linear(hidden_size, output_size)
        if sigma_type == 'fixed':
            self.register_buffer('sigma', torch.full((1, output_size), fixed_sigma_value))
        elif sigma_type in ('independent', 'happo'):
            self.log_sigma_param = nn.Parameter(torch.zeros(1, output_size))
        elif sigma_type == 'conditioned':
            self.log_sigma_layer = nn.Linear(hidden_size, output_size)

    def forward(self, x: torch.Tensor) -> Dict:
        x = self.main(x)
        mu = self.mu(x)
        if self.bound_type == 'tanh':
            mu = torch.tanh(mu)
        if self.sigma_type == 'fixed':
            sigma = self.sigma.expand(*mu.shape)
        elif self.sigma_type in ('independent', 'happo'):
            sigma = torch.exp(self.log_sigma_param).expand(*mu.shape)
        else:
            log_sigma = self.log_sigma_layer(x).clamp(-20, 2)
            sigma = torch.exp(log_sigma)
        return {'mu': mu, 'sigma': sigma}


class PopArtVHead(nn.Module):
    """Value head with PopArt normalization: {'pred','unnormalized_pred'}."""

    def __init__(
        self,
        hidden_size: int,
        output_size: int = 1,
        layer_num: int = 1,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        self.main = nn.Sequential(
            *[fc_block(hidden_size, hidden_size, activation=activation, norm_type=norm_type) for _ in range(layer_num)]
        )
        self.popart = PopArt(hidden_size, output_size)

    def forward(self, x: torch.Tensor) -> Dict:
        return self.popart(self.main(x))


class AttentionPolicyHead(nn.Module):
    """Dot-product pointer head: score keys by a query (variable action sets)."""

    def __init__(self):
        super().__init__()

    def forward(self, key: torch.Tensor, query: torch.Tensor) -> torch.Tensor:
        if query.dim() == 2 and key.dim() == 3:
            query = query.unsqueeze(1)
        logit = (key * query).sum(-1)
        return logit


class MultiHead(nn.Module):
    """One head per action dimension; outputs {'logit': [t1, t2, ...]}."""

    def __init__(self, head_cls: type, hidden_size: int, output_size_list: list, **head_kwargs):
        super().__init__()
        self.pred = nn.ModuleList([head_cls(hidden_size, size, **head_kwargs) for size in output_size_list])

    def forward(self, x: torch.Tensor) -> Dict:
        outputs = [h(x) for h in self.pred]
        merged = {}
        for k in outputs[0].keys():
            merged[k] = [o[k] for o in outputs]
        return merged


class EnsembleHead(nn.Module):
    """N independent Q(s,a) critics computed in one grouped conv1d (EDAC)."""

    def __init__(
        self,
        input_size: int,
        output_size: int,
        hidden_size: int,
        layer_num: int,
        ensemble_num: int,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        self.ensemble_num = ensemble_num
        from ding.torch_utils import conv1d_block
        layers = []
        dims = [input_size] + [hidden_size] * layer_num
        for i in range(layer_num):
            layers.append(
                conv1d_block(
                    dims[i] * ensemble_num, dims[i + 1] * ensemble_num, kernel_size=1, stride=1, groups=ensemble_num,
                    activation=activation, norm_type=norm_type
                )
            )
        layers.append(
            conv1d_block(hidden_size * ensemble_num, output_size * ensemble_num, 1, 1, groups=ensemble_num,
                         activation=None, norm_type=None)
        )
        self.pred = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> Dict:
        # x: [B, N*ensemble, 1]
        return {'pred': self.pred(x)}


head_cls_map = {
    'discrete': DiscreteHead,
    'dueling': DuelingHead,
    'distribution': DistributionHead,
    'rainbow': RainbowHead,
    'qrdqn': QRDQNHead,
    'quantile': QuantileHead,
    'fqf': FQFHead,
    'regression': RegressionHead,
    'reparameterization': ReparameterizationHead,
    'popart': PopArtVHead,
    'branching': BranchingHead,
    'attention_policy': AttentionPolicyHead,
    'multi': MultiHead,
    'ensemble': EnsembleHead,
    'sdn': StochasticDuelingHead,
}


def independent_normal_dist(logits) -> torch.distributions.Distribution:
    """[mu, sigma] pair or {'mu','sigma'} dict -> Independent Normal over the
    last dim (reference common/head.py:1440)."""
    if isinstance(logits, (list, tuple)):
        return torch.distributions.Independent(torch.distributions.Normal(*logits), 1)
    if isinstance(logits, dict):
        return torch.distributions.Independent(torch.distributions.Normal(logits['mu'], logits['sigma']), 1)
    raise TypeError(f"invalid logits type: {type(logits)}")
