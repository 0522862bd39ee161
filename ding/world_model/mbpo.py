"""MBPO ensemble dynamics model.

Parity: reference ding/world_model/mbpo.py:15 (EnsembleDynamicsModel with
gaussian heads, elite selection).
"""
import copy
import itertools
from typing import Tuple

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.utils import WORLD_MODEL_REGISTRY, EasyDict
from ding.torch_utils import unsqueeze_repeat
from .base_world_model import DynaWorldModel


class EnsembleFC(nn.Module):
    """Parallel linear layers for N ensemble members: [E, B, in] -> [E, B, out]."""

    def __init__(self, in_features: int, out_features: int, ensemble_size: int, weight_decay: float = 0.0):
        super().__init__()
        self.weight = nn.Parameter(torch.randn(ensemble_size, in_features, out_features) * 0.02)
        self.bias = nn.Parameter(torch.zeros(ensemble_size, 1, out_features))
        self.weight_decay = weight_decay

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.bmm(x, self.weight) + self.bias


class EnsembleModel(nn.Module):
    """Gaussian dynamics ensemble predicting (delta_obs, reward)."""

    def __init__(self, state_size, action_size, ensemble_size=7, hidden_size=200, learning_rate=1e-3):
        super().__init__()
        self.state_size = state_size
        self.action_size = action_size
        self.output_dim = state_size + 1  # delta_obs + reward
        self.ensemble_size = ensemble_size
        self.nn1 = EnsembleFC(state_size + action_size, hidden_size, ensemble_size, 2.5e-5)
        self.nn2 = EnsembleFC(hidden_size, hidden_size, ensemble_size, 5e-5)
        self.nn3 = EnsembleFC(hidden_size, hidden_size, ensemble_size, 7.5e-5)
        self.nn4 = EnsembleFC(hidden_size, 2 * self.output_dim, ensemble_size, 1e-4)
        self.max_logvar = nn.Parameter(torch.ones(1, self.output_dim) * 0.5)
        self.min_logvar = nn.Parameter(-torch.ones(1, self.output_dim) * 10)
        self.swish = nn.SiLU()
        self.optimizer = torch.optim.Adam(self.parameters(), lr=learning_rate)

    def forward(self, x: torch.Tensor, ret_log_var: bool = False):
        h = self.swish(self.nn1(x))
        h = self.swish(self.nn2(h))
        h = self.swish(self.nn3(h))
        out = self.nn4(h)
        mean = out[..., :self.output_dim]
        logvar = self.max_logvar - F.softplus(self.max_logvar - out[..., self.output_dim:])
        logvar = self.min_logvar + F.softplus(logvar - self.min_logvar)
        if ret_log_var:
            return mean, logvar
        return mean, torch.exp(logvar)

    def loss(self, mean, logvar, labels):
        inv_var = torch.exp(-logvar)
        mse = ((mean - labels).pow(2) * inv_var).mean(dim=(1, 2))
        var_loss = logvar.mean(dim=(1, 2))
        total = mse.sum() + var_loss.sum()
        total = total + 0.01 * self.max_logvar.sum() - 0.01 * self.min_logvar.sum()
        decay = sum(l.weight_decay * (l.weight ** 2).sum() / 2.0 for l in (self.nn1, self.nn2, self.nn3, self.nn4))
        return total + decay, mse.detach()


@WORLD_MODEL_REGISTRY.register('mbpo')
class MBPOWorldModel(DynaWorldModel):

    config = dict(
        model=dict(
            ensemble_size=7,
            elite_size=5,
            state_size=3,
            action_size=1,
            hidden_size=128,
            batch_size=256,
            max_epochs_since_update=5,
            deterministic_rollout=False,
        ),
    )

    def __init__(self, cfg: EasyDict, env=None, tb_logger=None):
        super().__init__(cfg, env, tb_logger)
        m = self.cfg.model
        self.ensemble_size = m.ensemble_size
        self.elite_size = m.elite_size
        self.model = EnsembleModel(
            m.state_size, m.action_size, m.ensemble_size, m.hidden_size
        ).to(self.device)
        self.elite_idx = list(range(self.elite_size))

    def train(self, env_buffer, envstep: int, train_iter: int) -> None:
        self.last_train_step = envstep
        n = min(env_buffer.count(), 4096)
        data = env_buffer.sample(n, train_iter)
        if not data:
            return
        obs = torch.stack([torch.as_tensor(d['obs'], dtype=torch.float32) for d in data]).to(self.device)
        action = torch.stack([torch.as_tensor(d['action'], dtype=torch.float32).reshape(-1) for d in data]).to(self.device)
        next_obs = torch.stack([torch.as_tensor(d['next_obs'], dtype=torch.float32) for d in data]).to(self.device)
        reward = torch.stack([torch.as_tensor(d['reward'], dtype=torch.float32).reshape(-1)[0:1] for d in data]).to(self.device)
        inputs = torch.cat([obs, action], dim=-1)
        labels = torch.cat([next_obs - obs, reward], dim=-1)
        E = self.ensemble_size
        B = inputs.shape[0]
        bs = self.cfg.model.batch_size
        for start in range(0, B, bs):
            x = inputs[start:start + bs]
            y = labels[start:start + bs]
            # bootstrap resample per member
            idx = torch.randint(0, x.shape[0], (E, x.shape[0]), device=self.device)
            xe = x[idx]
            ye = y[idx]
            mean, logvar = self.model(xe, ret_log_var=True)
            loss, mse = self.model.loss(mean, logvar, ye)
            self.model.optimizer.zero_grad()
            loss.backward()
            self.model.optimizer.step()
        # elite selection by per-member mse
        with torch.no_grad():
            mean, logvar = self.model(unsqueeze_repeat(inputs, E), ret_log_var=True)
            per_member_mse = (mean - unsqueeze_repeat(labels, E)).pow(2).mean(dim=(1, 2))
            self.elite_idx = per_member_mse.argsort()[:self.elite_size].tolist()

    def eval(self, env_buffer, envstep: int, train_iter: int) -> None:
        self.last_eval_step = envstep

    def step(self, obs: torch.Tensor, action: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        x = torch.cat([obs, action.float()], dim=-1)
        E = self.ensemble_size
        with torch.no_grad():
            mean, var = self.model(unsqueeze_repeat(x, E))
            if self.cfg.model.deterministic_rollout:
                sample = mean
            else:
                sample = mean + var.sqrt() * torch.randn_like(mean)
            # pick a random elite member per sample
            B = obs.shape[0]
            member = torch.tensor(
                np.random.choice(self.elite_idx, B), device=obs.device, dtype=torch.long
            )
            sample = sample[member, torch.arange(B, device=obs.device)]
            delta, reward = sample[..., :-1], sample[..., -1]
            next_obs = obs + delta
            done = torch.zeros(B, dtype=torch.bool, device=obs.device)
        return reward, next_obs, done
