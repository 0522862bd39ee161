"""PopArt value head: adaptive target normalization with output-preserving
weight updates (van Hasselt et al. 2016).

Parity: reference ding/torch_utils/network/popart.py (PopArt).
"""
import math

import torch
import torch.nn as nn


class PopArt(nn.Module):

    def __init__(self, input_features: int, output_features: int = 1, beta: float = 0.5):
        super().__init__()
        self.beta = beta
        self.input_features = input_features
        self.output_features = output_features
        self.weight = nn.Parameter(torch.empty(output_features, input_features))
        self.bias = nn.Parameter(torch.zeros(output_features))
        self.register_buffer('mu', torch.zeros(output_features))
        self.register_buffer('sigma', torch.ones(output_features))
        self.register_buffer('v', torch.ones(output_features))  # second moment
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))

    def forward(self, x: torch.Tensor) -> dict:
        normalized = torch.nn.functional.linear(x, self.weight, self.bias)
        with torch.no_grad():
            output = normalized * self.sigma + self.mu
        return {'pred': normalized, 'unnormalized_pred': output}

    def update_parameters(self, value: torch.Tensor) -> dict:
        """Update (mu, sigma) from a batch of targets; rescale weight/bias so
        unnormalized outputs are unchanged."""
        old_mu, old_sigma = self.mu.clone(), self.sigma.clone()
        batch_mu = value.mean(dim=tuple(range(value.dim() - 1))) if value.dim() > 1 else value.mean()
        batch_v = (value ** 2).mean(dim=tuple(range(value.dim() - 1))) if value.dim() > 1 else (value ** 2).mean()
        self.mu = (1 - self.beta) * self.mu + self.beta * batch_mu.reshape_as(self.mu)
        self.v = (1 - self.beta) * self.v + self.beta * batch_v.reshape_as(self.v)
        self.sigma = torch.sqrt((self.v - self.mu ** 2).clamp(min=1e-4)).clamp(1e-4, 1e6)
        with torch.no_grad():
            self.weight.data = self.weight.data * (old_sigma / self.sigma).unsqueeze(-1)
            self.bias.data = (old_sigma * self.bias.data + old_mu - self.mu) / self.sigma
        return {'new_mean': self.mu, 'new_std': self.sigma}

    def normalize(self, x: torch.Tensor) -> torch.Tensor:
        return (x - self.mu) / self.sigma

    def unnormalize(self, x: torch.Tensor) -> torch.Tensor:
        return x * self.sigma + self.mu
