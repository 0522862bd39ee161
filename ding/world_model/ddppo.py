"""DDPPO world model: gradient-through-dynamics variant of the MBPO
ensemble (differentiable step()).

Parity: reference ding/world_model/ddppo.py:80.
"""
from typing import Tuple

import numpy as np
import torch

from ding.utils import WORLD_MODEL_REGISTRY, EasyDict
from ding.torch_utils import unsqueeze_repeat
from .base_world_model import DreamWorldModel
from .mbpo import MBPOWorldModel


@WORLD_MODEL_REGISTRY.register('ddppo')
class DDPPOWorldMode(MBPOWorldModel, DreamWorldModel):
    """Same ensemble as MBPO but step() keeps the graph so policy gradients
    flow through imagined transitions."""

    def step(
        self, obs: torch.Tensor, action: torch.Tensor, keep_ensemble: bool = False
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        E = self.ensemble_size
        if keep_ensemble:
            # STEVE lane: obs [E, B, O] — each member propagates its own row
            x = torch.cat([obs, action.float()], dim=-1)
            mean, var = self.model(x)
            sample = mean + var.sqrt() * torch.randn_like(mean)
            delta, reward = sample[..., :-1], sample[..., -1]
            next_obs = obs + delta
            done = torch.zeros(obs.shape[:-1], dtype=torch.bool, device=obs.device)
            return reward, next_obs, done
        x = torch.cat([obs, action.float()], dim=-1)
        mean, var = self.model(unsqueeze_repeat(x, E))
        sample = mean + var.sqrt() * torch.randn_like(mean)
        B = obs.shape[0]
        member = torch.tensor(np.random.choice(self.elite_idx, B), device=obs.device, dtype=torch.long)
        sample = sample[member, torch.arange(B, device=obs.device)]
        delta, reward = sample[..., :-1], sample[..., -1]
        next_obs = obs + delta
        done = torch.zeros(B, dtype=torch.bool, device=obs.device)
        return reward, next_obs, done
