"""NGU intrinsic rewards: lifelong RND modulator + episodic embedding-kNN
novelty.

Parity: reference ding/reward_model/ngu_reward_model.py (RndNGURewardModel:97,
EpisodicNGURewardModel:245).
"""
import copy
import numpy as np
from collections import defaultdict
from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.utils import REWARD_MODEL_REGISTRY, EasyDict, RunningMeanStd
from .base_reward_model import BaseRewardModel
from .exploration import RndNetwork, _obs_encoder


@REWARD_MODEL_REGISTRY.register('rnd-ngu')
class RndNGURewardModel(BaseRewardModel):
    """Lifelong novelty: RND error normalized to a multiplier alpha."""

    config = dict(
        type='rnd-ngu',
        obs_shape=4,
        hidden_size_list=[64, 64],
        learning_rate=1e-3,
        batch_size=64,
        update_per_collect=10,
    )

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.reward_model = RndNetwork(self.cfg.obs_shape, self.cfg.hidden_size_list).to(device)
        self.opt = torch.optim.Adam(self.reward_model.predictor.parameters(), lr=self.cfg.learning_rate)
        self.train_obs = []
        self._rms = RunningMeanStd(epsilon=1e-4)

    def collect_data(self, data: list) -> None:
        for d in data:
            obs = d['obs']
            if isinstance(obs, (list, tuple)):  # r2d2 unroll sample: [T] of obs
                self.train_obs.extend(obs)
            elif isinstance(obs, torch.Tensor) and obs.dim() == 2:  # [T, obs]
                self.train_obs.extend(obs.unbind(0))
            else:
                self.train_obs.append(obs)

    def clear_data(self) -> None:
        self.train_obs = []

    def train(self, data=None) -> None:
        if not self.train_obs:
            return
        for _ in range(self.cfg.update_per_collect):
            idx = torch.randint(0, len(self.train_obs), (min(self.cfg.batch_size, len(self.train_obs)), ))
            obs = torch.stack([torch.as_tensor(self.train_obs[i], dtype=torch.float32) for i in idx]).to(self.device)
            p, t = self.reward_model(obs)
            loss = F.mse_loss(p, t.detach())
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    @staticmethod
    def _sample_obs(d):
        """One representative obs per sample: sequences (r2d2 unrolls) use the
        mean over the window."""
        obs = d['obs']
        if isinstance(obs, (list, tuple)):
            return torch.stack([torch.as_tensor(o, dtype=torch.float32) for o in obs]).mean(0)
        obs = torch.as_tensor(obs, dtype=torch.float32)
        return obs.mean(0) if obs.dim() == 2 else obs

    def estimate(self, data: list) -> torch.Tensor:
        """Return the alpha multiplier per transition (not reward rewrite)."""
        obs = torch.stack([self._sample_obs(d) for d in data]).to(self.device)
        with torch.no_grad():
            p, t = self.reward_model(obs)
            err = (p - t).pow(2).sum(1)
        self._rms.update(err.cpu().numpy().reshape(-1, 1))
        alpha = 1 + (err - float(np.asarray(self._rms.mean).reshape(-1)[0])) / (float(np.asarray(self._rms.std).reshape(-1)[0]) + 1e-8)
        return alpha.clamp(1.0, 5.0)


@REWARD_MODEL_REGISTRY.register('episodic')
class EpisodicNGURewardModel(BaseRewardModel):
    """Episodic novelty: controllable-state embedding + kNN pseudo-count."""

    config = dict(
        type='episodic',
        obs_shape=4,
        action_shape=2,
        hidden_size_list=[64, 64],
        learning_rate=1e-3,
        batch_size=64,
        update_per_collect=10,
        k=10,
        kernel_eps=1e-3,
    )

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.embed = _obs_encoder(self.cfg.obs_shape, self.cfg.hidden_size_list).to(device)
        feat = self.cfg.hidden_size_list[-1]
        self.inverse = nn.Sequential(nn.Linear(feat * 2, 64), nn.ReLU(), nn.Linear(64, self.cfg.action_shape)).to(device)
        self.opt = torch.optim.Adam(list(self.embed.parameters()) + list(self.inverse.parameters()),
                                    lr=self.cfg.learning_rate)
        self.train_data = []
        self._episode_memory = defaultdict(list)  # env_id -> embeddings

    def collect_data(self, data: list) -> None:
        for d in data:
            obs = d['obs']
            if isinstance(obs, (list, tuple)) or (isinstance(obs, torch.Tensor) and obs.dim() == 2):
                # r2d2 unroll sample: explode the [T] sequence into transitions
                T = len(obs)
                next_obs = d.get('next_obs')  # r2d2 unrolls have none: shift obs
                for t in range(T):
                    nxt = next_obs[t] if next_obs is not None else obs[min(t + 1, T - 1)]
                    self.train_data.append({
                        'obs': obs[t],
                        'next_obs': nxt,
                        'action': d['action'][t],
                        'env_id': d.get('env_id', 0),
                    })
            else:
                self.train_data.append(d)

    def clear_data(self) -> None:
        self.train_data = []

    def train(self, data=None) -> None:
        if not self.train_data:
            return
        for _ in range(self.cfg.update_per_collect):
            idx = torch.randint(0, len(self.train_data), (min(self.cfg.batch_size, len(self.train_data)), ))
            batch = [self.train_data[i] for i in idx]
            obs = torch.stack([torch.as_tensor(b['obs'], dtype=torch.float32) for b in batch]).to(self.device)
            next_obs = torch.stack([torch.as_tensor(b['next_obs'], dtype=torch.float32) for b in batch]).to(self.device)
            action = torch.stack([torch.as_tensor(b['action']).reshape(()) for b in batch]).long().to(self.device)
            f1, f2 = self.embed(obs), self.embed(next_obs)
            logit = self.inverse(torch.cat([f1, f2], dim=1))
            loss = F.cross_entropy(logit, action)
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    def reset_episode(self, env_id: int) -> None:
        self._episode_memory[env_id] = []

    def estimate(self, data: list) -> torch.Tensor:
        """Per-transition episodic novelty reward 1/sqrt(N_knn)."""
        rewards = []
        with torch.no_grad():
            for d in data:
                env_id = d.get('env_id', 0)
                if isinstance(env_id, torch.Tensor):
                    env_id = int(env_id.reshape(-1)[0])
                env_id = int(env_id)
                obs = RndNGURewardModel._sample_obs(d).unsqueeze(0).to(self.device)
                e = self.embed(obs).squeeze(0)
                mem = self._episode_memory[env_id]
                if len(mem) == 0:
                    r = 1.0
                else:
                    M = torch.stack(mem)
                    dists = (M - e).pow(2).sum(1)
                    k = min(self.cfg.k, len(mem))
                    topk = dists.topk(k, largest=False).values
                    dm = topk.mean().clamp(min=1e-8)
                    kernel = self.cfg.kernel_eps / (topk / dm + self.cfg.kernel_eps)
                    r = float(1.0 / (kernel.sum().sqrt() + 1e-3))
                mem.append(e)
                rewards.append(r)
        return torch.tensor(rewards)


def fusion_reward(data: list, episodic_reward: torch.Tensor, alpha: torch.Tensor, beta: float = 0.3) -> list:
    """NGU reward fusion: r_total = r_ext + beta * r_episodic * alpha."""
    out = []
    for d, er, a in zip(data, episodic_reward, alpha):
        nd = dict(d)
        intrinsic = beta * float(er) * float(a)
        nd['reward'] = torch.as_tensor(d['reward'], dtype=torch.float32) + intrinsic
        out.append(nd)
    return out
