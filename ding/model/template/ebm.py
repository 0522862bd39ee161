"""Energy-based model + stochastic optimizers for implicit BC.

Parity: reference ding/model/template/ebm.py ('ebm' with DFO/MCMC samplers).
"""
from typing import Dict, Optional

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ding.torch_utils import MLP


@MODEL_REGISTRY.register('ebm')
class EBM(nn.Module):
    """E(s, a) scalar energy."""

    def __init__(self, obs_shape: int, action_shape: int, hidden_size: int = 512, hidden_layer_num: int = 4,
                 **kwargs):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        self.net = MLP(obs_shape + action_shape, hidden_size, 1, hidden_layer_num, activation='relu',
                       output_activation=False, output_norm=False)

    def forward(self, obs: torch.Tensor, action: torch.Tensor) -> torch.Tensor:
        """obs [B, N, O], action [B, N, A] -> energy [B, N]."""
        x = torch.cat([obs, action], dim=-1)
        return self.net(x).squeeze(-1)


class DFO:
    """Derivative-free optimizer: iterated random shrinking search."""

    def __init__(self, noise_scale: float = 0.33, noise_shrink: float = 0.5, iters: int = 3,
                 train_samples: int = 8, inference_samples: int = 512, bounds=(-1.0, 1.0)):
        self.noise_scale = noise_scale
        self.noise_shrink = noise_shrink
        self.iters = iters
        self.train_samples = train_samples
        self.inference_samples = inference_samples
        self.bounds = bounds

    def sample(self, obs: torch.Tensor, ebm: EBM, action_dim: int, n: Optional[int] = None) -> torch.Tensor:
        """Negatives for InfoNCE training: uniform samples [B, n, A]."""
        n = n or self.train_samples
        B = obs.shape[0]
        lo, hi = self.bounds
        return torch.empty(B, n, action_dim, device=obs.device).uniform_(lo, hi)

    def infer(self, obs: torch.Tensor, ebm: EBM, action_dim: int) -> torch.Tensor:
        """argmin_a E(s, a) via shrinking random search. obs [B, O] -> [B, A]."""
        B = obs.shape[0]
        lo, hi = self.bounds
        n = self.inference_samples
        samples = torch.empty(B, n, action_dim, device=obs.device).uniform_(lo, hi)
        noise = self.noise_scale
        obs_tiled = obs.unsqueeze(1).expand(B, n, obs.shape[-1])
        for _ in range(self.iters):
            energy = ebm(obs_tiled, samples)  # [B, n]
            prob = torch.softmax(-energy, dim=-1)
            idx = torch.multinomial(prob, n, replacement=True)
            samples = samples.gather(1, idx.unsqueeze(-1).expand(B, n, action_dim))
            samples = samples + noise * torch.randn_like(samples)
            samples = samples.clamp(lo, hi)
            noise *= self.noise_shrink
        energy = ebm(obs_tiled, samples)
        best = energy.argmin(dim=-1)
        return samples[torch.arange(B, device=obs.device), best]


class LangevinMCMC(DFO):
    """Gradient-based Langevin sampler."""

    def __init__(self, step_size: float = 0.1, iters: int = 20, **kwargs):
        super().__init__(**kwargs)
        self.step_size = step_size
        self.mcmc_iters = iters

    def infer(self, obs: torch.Tensor, ebm: EBM, action_dim: int) -> torch.Tensor:
        B = obs.shape[0]
        lo, hi = self.bounds
        a = torch.empty(B, action_dim, device=obs.device).uniform_(lo, hi).requires_grad_(True)
        for _ in range(self.mcmc_iters):
            energy = ebm(obs.unsqueeze(1), a.unsqueeze(1)).sum()
            grad = torch.autograd.grad(energy, a)[0]
            with torch.no_grad():
                a = a - 0.5 * self.step_size * grad + (self.step_size ** 0.5) * 0.1 * torch.randn_like(a)
                a = a.clamp(lo, hi)
            a = a.requires_grad_(True)
        return a.detach()
