"""Imitation / IRL reward models: GAIL, guided cost (MaxEnt IOC), PWIL, RED,
PDEIL, TREX/DREX (preference-based).

Parity: reference ding/reward_model/{gail_irl_model.py:106,
guided_cost_reward_model.py:36, pwil_irl_model.py, red_irl_model.py,
pdeil_irl_model.py, trex_reward_model.py:132, drex_reward_model.py:10}.
"""
import copy
import pickle
import random
from typing import Any, List

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.utils import REWARD_MODEL_REGISTRY, EasyDict
from .base_reward_model import BaseRewardModel




def _rewrite_reward(item: dict, value: float) -> None:
    """Overwrite the reward while preserving its original shape (n-step
    reward vectors stay n-step vectors)."""
    old = torch.as_tensor(item['reward'], dtype=torch.float32)
    if old.dim() == 0:
        old = old.reshape(1)
    item['reward'] = torch.full_like(old, float(value))

def _load_expert(path: str) -> list:
    with open(path, 'rb') as f:
        return pickle.load(f)


@REWARD_MODEL_REGISTRY.register('gail')
class GailRewardModel(BaseRewardModel):
    """Discriminator D(s, a); reward = -log(1 - D) (non-saturating)."""

    config = dict(
        type='gail',
        input_size=5,
        hidden_size=64,
        batch_size=64,
        learning_rate=1e-3,
        update_per_collect=100,
        expert_data_path='expert.pkl',
        action_size=1,
        collect_count=1000,
    )

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.reward_model = nn.Sequential(
            nn.Linear(self.cfg.input_size, self.cfg.hidden_size), nn.Tanh(),
            nn.Linear(self.cfg.hidden_size, 1)
        ).to(device)
        self.opt = torch.optim.Adam(self.reward_model.parameters(), lr=self.cfg.learning_rate)
        self.train_data = []
        self.expert_data = []

    def load_expert_data(self, data=None) -> None:
        if data is not None:
            self.expert_data = data
        else:
            self.expert_data = _load_expert(self.cfg.expert_data_path)

    def _concat(self, item) -> torch.Tensor:
        obs = torch.as_tensor(item['obs'], dtype=torch.float32).reshape(-1)
        act = torch.as_tensor(item['action'], dtype=torch.float32).reshape(-1)
        return torch.cat([obs, act])

    def collect_data(self, data: list) -> None:
        self.train_data.extend(data)

    def clear_data(self) -> None:
        self.train_data = []

    def train(self, data=None) -> None:
        if not self.train_data or not self.expert_data:
            return
        for _ in range(self.cfg.update_per_collect):
            agent_batch = random.sample(self.train_data, min(self.cfg.batch_size, len(self.train_data)))
            expert_batch = random.sample(self.expert_data, min(self.cfg.batch_size, len(self.expert_data)))
            agent_x = torch.stack([self._concat(b) for b in agent_batch]).to(self.device)
            expert_x = torch.stack([self._concat(b) for b in expert_batch]).to(self.device)
            agent_logit = self.reward_model(agent_x)
            expert_logit = self.reward_model(expert_x)
            loss = F.binary_cross_entropy_with_logits(expert_logit, torch.ones_like(expert_logit)) + \
                F.binary_cross_entropy_with_logits(agent_logit, torch.zeros_like(agent_logit))
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    def estimate(self, data: list) -> List[dict]:
        out = self.reward_deepcopy(data)
        x = torch.stack([self._concat(d) for d in out]).to(self.device)
        with torch.no_grad():
            d_prob = torch.sigmoid(self.reward_model(x)).squeeze(-1)
            reward = -torch.log(1 - d_prob + 1e-8)
        for item, r in zip(out, reward):
            _rewrite_reward(item, r.item())
        return out


@REWARD_MODEL_REGISTRY.register('guided_cost')
class GuidedCostRewardModel(BaseRewardModel):
    """MaxEnt IOC cost net trained against expert demos (GCL)."""

    config = dict(
        type='guided_cost',
        input_size=5,
        hidden_size=64,
        batch_size=64,
        learning_rate=1e-3,
        update_per_collect=100,
        log_every_n_train=50,
    )

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.reward_model = nn.Sequential(
            nn.Linear(self.cfg.input_size, self.cfg.hidden_size), nn.ReLU(),
            nn.Linear(self.cfg.hidden_size, 1)
        ).to(device)
        self.opt = torch.optim.Adam(self.reward_model.parameters(), lr=self.cfg.learning_rate)
        self.expert_data = []
        self.train_data = []

    def load_expert_data(self, data=None) -> None:
        self.expert_data = data or []

    def collect_data(self, data: list) -> None:
        self.train_data.extend(data)

    def clear_data(self) -> None:
        self.train_data = []

    def _x(self, item):
        obs = torch.as_tensor(item['obs'], dtype=torch.float32).reshape(-1)
        act = torch.as_tensor(item['action'], dtype=torch.float32).reshape(-1)
        return torch.cat([obs, act])

    def train(self, expert_demo=None, samp=None) -> None:
        expert = expert_demo or self.expert_data
        agent = samp or self.train_data
        if not expert or not agent:
            return
        for _ in range(self.cfg.update_per_collect):
            e = torch.stack([self._x(b) for b in random.sample(expert, min(self.cfg.batch_size, len(expert)))])
            a = torch.stack([self._x(b) for b in random.sample(agent, min(self.cfg.batch_size, len(agent)))])
            e, a = e.to(self.device), a.to(self.device)
            cost_e = self.reward_model(e)
            cost_a = self.reward_model(a)
            # IOC objective: expert cost low, log-sum-exp of sample costs as partition
            loss = cost_e.mean() + torch.logsumexp(-cost_a, dim=0).mean()
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    def estimate(self, data: list) -> List[dict]:
        out = self.reward_deepcopy(data)
        x = torch.stack([self._x(d) for d in out]).to(self.device)
        with torch.no_grad():
            reward = -self.reward_model(x).squeeze(-1)
        for item, r in zip(out, reward):
            _rewrite_reward(item, r.item())
        return out


@REWARD_MODEL_REGISTRY.register('pwil')
class PwilRewardModel(BaseRewardModel):
    """Primal Wasserstein imitation: greedy transport cost to expert set."""

    config = dict(type='pwil', sample_size=500, alpha=5.0, beta=5.0)

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.expert_data = []
        self.train_data = []

    def load_expert_data(self, data=None) -> None:
        self.expert_data = data or []

    def collect_data(self, data: list) -> None:
        self.train_data.extend(data)

    def clear_data(self) -> None:
        self.train_data = []

    def train(self, data=None) -> None:
        pass  # non-parametric

    def estimate(self, data: list) -> List[dict]:
        out = self.reward_deepcopy(data)
        if not self.expert_data:
            return out
        exp = torch.stack([
            torch.cat([
                torch.as_tensor(d['obs'], dtype=torch.float32).reshape(-1),
                torch.as_tensor(d['action'], dtype=torch.float32).reshape(-1)
            ]) for d in self.expert_data
        ])
        for item in out:
            x = torch.cat([
                torch.as_tensor(item['obs'], dtype=torch.float32).reshape(-1),
                torch.as_tensor(item['action'], dtype=torch.float32).reshape(-1)
            ])
            dist = (exp - x).norm(dim=1).min()
            reward = self.cfg.alpha * torch.exp(-self.cfg.beta * dist)
            _rewrite_reward(item, reward.item())
        return out


@REWARD_MODEL_REGISTRY.register('red')
class RedRewardModel(BaseRewardModel):
    """Random expert distillation: RND trained only on expert data."""

    config = dict(type='red', input_size=5, hidden_size=64, batch_size=64, learning_rate=1e-3,
                  update_per_collect=100, sigma=0.5)

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.target = nn.Sequential(nn.Linear(self.cfg.input_size, self.cfg.hidden_size), nn.ReLU(),
                                    nn.Linear(self.cfg.hidden_size, 32)).to(device)
        self.predictor = nn.Sequential(nn.Linear(self.cfg.input_size, self.cfg.hidden_size), nn.ReLU(),
                                       nn.Linear(self.cfg.hidden_size, 32)).to(device)
        for p in self.target.parameters():
            p.requires_grad = False
        self.opt = torch.optim.Adam(self.predictor.parameters(), lr=self.cfg.learning_rate)
        self.expert_data = []

    def load_expert_data(self, data=None) -> None:
        self.expert_data = data or []

    def collect_data(self, data) -> None:
        pass

    def clear_data(self) -> None:
        pass

    def _x(self, item):
        return torch.cat([
            torch.as_tensor(item['obs'], dtype=torch.float32).reshape(-1),
            torch.as_tensor(item['action'], dtype=torch.float32).reshape(-1)
        ])

    def train(self, data=None) -> None:
        if not self.expert_data:
            return
        for _ in range(self.cfg.update_per_collect):
            batch = random.sample(self.expert_data, min(self.cfg.batch_size, len(self.expert_data)))
            x = torch.stack([self._x(b) for b in batch]).to(self.device)
            loss = F.mse_loss(self.predictor(x), self.target(x).detach())
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    def estimate(self, data: list) -> List[dict]:
        out = self.reward_deepcopy(data)
        x = torch.stack([self._x(d) for d in out]).to(self.device)
        with torch.no_grad():
            err = (self.predictor(x) - self.target(x)).pow(2).mean(1)
            reward = torch.exp(-self.cfg.sigma * err)
        for item, r in zip(out, reward):
            _rewrite_reward(item, r.item())
        return out


@REWARD_MODEL_REGISTRY.register('pdeil')
class PdeilRewardModel(BaseRewardModel):
    """Probability-density estimation IL: gaussian density ratio reward."""

    config = dict(type='pdeil', alpha=0.5, discrete_action=False)

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.expert_data = []
        self._mean = None
        self._cov_inv = None

    def load_expert_data(self, data=None) -> None:
        self.expert_data = data or []

    def collect_data(self, data) -> None:
        pass

    def clear_data(self) -> None:
        pass

    def train(self, data=None) -> None:
        if not self.expert_data:
            return
        obs = torch.stack([torch.as_tensor(d['obs'], dtype=torch.float32).reshape(-1) for d in self.expert_data])
        self._mean = obs.mean(0)
        cov = torch.from_numpy(np.cov(obs.numpy(), rowvar=False)).float()
        self._cov_inv = torch.linalg.pinv(cov + 1e-4 * torch.eye(cov.shape[0]))

    def estimate(self, data: list) -> List[dict]:
        out = self.reward_deepcopy(data)
        if self._mean is None:
            return out
        for item in out:
            x = torch.as_tensor(item['obs'], dtype=torch.float32).reshape(-1)
            d = x - self._mean
            maha = (d @ self._cov_inv @ d).clamp(min=0)
            _rewrite_reward(item, torch.exp(-0.5 * maha).item())
        return out


class TrexNetwork(nn.Module):

    def __init__(self, input_size: int, hidden_size: int = 64):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(input_size, hidden_size), nn.ReLU(), nn.Linear(hidden_size, hidden_size), nn.ReLU(),
            nn.Linear(hidden_size, 1)
        )

    def forward(self, traj: torch.Tensor) -> torch.Tensor:
        return self.net(traj).sum(dim=-2)  # sum per-step rewards over the trajectory


@REWARD_MODEL_REGISTRY.register('trex')
class TrexRewardModel(BaseRewardModel):
    """Preference-based reward from ranked trajectories (Bradley-Terry)."""

    config = dict(
        type='trex',
        input_size=4,
        hidden_size=64,
        batch_size=64,
        learning_rate=1e-4,
        update_per_collect=100,
        num_snippets=100,
        min_snippet_length=5,
        max_snippet_length=20,
    )

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.reward_model = TrexNetwork(self.cfg.input_size, self.cfg.hidden_size).to(device)
        self.opt = torch.optim.Adam(self.reward_model.parameters(), lr=self.cfg.learning_rate)
        self.ranked_trajectories: List[list] = []  # worst -> best

    def load_ranked_trajectories(self, trajs: List[list]) -> None:
        self.ranked_trajectories = trajs

    def collect_data(self, data) -> None:
        pass

    def clear_data(self) -> None:
        self.ranked_trajectories = []

    def _snippet(self, traj: list) -> torch.Tensor:
        L = random.randint(self.cfg.min_snippet_length, min(self.cfg.max_snippet_length, len(traj)))
        start = random.randint(0, len(traj) - L)
        return torch.stack([
            torch.as_tensor(t['obs'], dtype=torch.float32).reshape(-1) for t in traj[start:start + L]
        ])

    def train(self, data=None) -> None:
        if len(self.ranked_trajectories) < 2:
            return
        n = len(self.ranked_trajectories)
        for _ in range(self.cfg.update_per_collect):
            i, j = sorted(random.sample(range(n), 2))
            worse, better = self.ranked_trajectories[i], self.ranked_trajectories[j]
            if len(worse) < self.cfg.min_snippet_length or len(better) < self.cfg.min_snippet_length:
                continue
            r_worse = self.reward_model(self._snippet(worse).to(self.device))
            r_better = self.reward_model(self._snippet(better).to(self.device))
            logits = torch.stack([r_worse, r_better]).reshape(1, 2)
            loss = F.cross_entropy(logits, torch.tensor([1], device=self.device))
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    def estimate(self, data: list) -> List[dict]:
        out = self.reward_deepcopy(data)
        obs = torch.stack([torch.as_tensor(d['obs'], dtype=torch.float32).reshape(-1) for d in out]).to(self.device)
        with torch.no_grad():
            reward = self.reward_model.net(obs).squeeze(-1)
        for item, r in zip(out, reward):
            _rewrite_reward(item, r.item())
        return out


@REWARD_MODEL_REGISTRY.register('drex')
class DrexRewardModel(TrexRewardModel):
    """DREX: TREX over noise-ranked BC rollouts (more noise = worse rank)."""

    config = dict(
        type='drex',
        input_size=4,
        hidden_size=64,
        batch_size=64,
        learning_rate=1e-4,
        update_per_collect=100,
        num_snippets=100,
        min_snippet_length=5,
        max_snippet_length=20,
        noise_levels=[0.0, 0.3, 0.6, 1.0],
    )
