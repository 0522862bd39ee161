"""Intrinsic-exploration reward models: RND, ICM.

Parity: reference ding/reward_model/rnd_reward_model.py:51 and
icm_reward_model.py:125.
"""
import copy
import numpy as np
from typing import Any, List, Optional, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.model import FCEncoder, ConvEncoder
from ding.utils import REWARD_MODEL_REGISTRY, EasyDict, RunningMeanStd
from .base_reward_model import BaseRewardModel


def _obs_encoder(obs_shape, hidden_size_list):
    if isinstance(obs_shape, int) or len(obs_shape) == 1:
        from ding.utils import squeeze
        return FCEncoder(squeeze(obs_shape), hidden_size_list)
    return ConvEncoder(obs_shape, hidden_size_list)


class RndNetwork(nn.Module):

    def __init__(self, obs_shape, hidden_size_list):
        super().__init__()
        self.target = _obs_encoder(obs_shape, hidden_size_list)
        self.predictor = _obs_encoder(obs_shape, hidden_size_list)
        for p in self.target.parameters():
            p.requires_grad = False

    def forward(self, obs: torch.Tensor):
        with torch.no_grad():
            t_feat = self.target(obs)
        p_feat = self.predictor(obs)
        return p_feat, t_feat


@REWARD_MODEL_REGISTRY.register('rnd')
class RndRewardModel(BaseRewardModel):

    config = dict(
        type='rnd',
        intrinsic_reward_type='add',
        learning_rate=1e-3,
        obs_shape=4,
        hidden_size_list=[64, 64],
        batch_size=64,
        update_per_collect=10,
        obs_norm=True,
        obs_norm_clamp_min=-1,
        obs_norm_clamp_max=1,
        intrinsic_reward_weight=0.01,
        extrinsic_reward_norm=True,
        extrinsic_reward_norm_max=1,
    )

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.reward_model = RndNetwork(self.cfg.obs_shape, self.cfg.hidden_size_list).to(device)
        self.opt = torch.optim.Adam(self.reward_model.predictor.parameters(), lr=self.cfg.learning_rate)
        self.train_obs = []
        self._running_mean_std_rnd_reward = RunningMeanStd(epsilon=1e-4)
        self._running_mean_std_rnd_obs = RunningMeanStd(epsilon=1e-4)
        assert self.cfg.intrinsic_reward_type in ('add', 'new', 'assign')

    def collect_data(self, data: list) -> None:
        self.train_obs.extend([item['obs'] for item in data])

    def clear_data(self) -> None:
        self.train_obs = []

    def _norm_obs(self, obs: torch.Tensor) -> torch.Tensor:
        if not self.cfg.obs_norm:
            return obs
        import numpy as np
        self._running_mean_std_rnd_obs.update(obs.detach().cpu().numpy().reshape(obs.shape[0], -1).mean(-1, keepdims=True))
        mean = float(np.asarray(self._running_mean_std_rnd_obs.mean).reshape(-1)[0])
        std = float(np.asarray(self._running_mean_std_rnd_obs.std).reshape(-1)[0])
        return ((obs - mean) / (std + 1e-8)).clamp(self.cfg.obs_norm_clamp_min, self.cfg.obs_norm_clamp_max)

    def train(self, data=None) -> None:
        if not self.train_obs:
            return
        for _ in range(self.cfg.update_per_collect):
            idx = torch.randint(0, len(self.train_obs), (min(self.cfg.batch_size, len(self.train_obs)), ))
            obs = torch.stack([torch.as_tensor(self.train_obs[i], dtype=torch.float32) for i in idx]).to(self.device)
            obs = self._norm_obs(obs)
            p_feat, t_feat = self.reward_model(obs)
            loss = F.mse_loss(p_feat, t_feat.detach())
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    def estimate(self, data: list) -> List[dict]:
        train_data = self.reward_deepcopy(data)
        obs = torch.stack([torch.as_tensor(item['obs'], dtype=torch.float32) for item in train_data]).to(self.device)
        obs = self._norm_obs(obs)
        with torch.no_grad():
            p_feat, t_feat = self.reward_model(obs)
            rnd_reward = (p_feat - t_feat).pow(2).sum(1)
            self._running_mean_std_rnd_reward.update(rnd_reward.cpu().numpy().reshape(-1, 1))
            rnd_reward = rnd_reward / float(np.asarray(self._running_mean_std_rnd_reward.std).reshape(-1)[0] + 1e-8)
        for item, ir in zip(train_data, rnd_reward):
            rew = item['reward']
            if self.cfg.extrinsic_reward_norm:
                rew = rew / self.cfg.extrinsic_reward_norm_max
            ir = ir.item() * self.cfg.intrinsic_reward_weight
            if self.cfg.intrinsic_reward_type == 'add':
                item['reward'] = rew + ir
            elif self.cfg.intrinsic_reward_type == 'new':
                item['intrinsic_reward'] = torch.as_tensor([ir])
            else:
                item['reward'] = torch.as_tensor([ir])
        return train_data


class IcmNetwork(nn.Module):
    """Feature encoder + forward model + inverse model."""

    def __init__(self, obs_shape, hidden_size_list, action_shape):
        super().__init__()
        self.feature = _obs_encoder(obs_shape, hidden_size_list)
        feat = hidden_size_list[-1]
        self.action_shape = action_shape
        self.forward_net = nn.Sequential(nn.Linear(feat + action_shape, 128), nn.ReLU(), nn.Linear(128, feat))
        self.inverse_net = nn.Sequential(nn.Linear(feat * 2, 128), nn.ReLU(), nn.Linear(128, action_shape))

    def forward(self, state, next_state, action_onehot):
        f1 = self.feature(state)
        f2 = self.feature(next_state)
        pred_next = self.forward_net(torch.cat([f1, action_onehot], 1))
        pred_action_logit = self.inverse_net(torch.cat([f1, f2], 1))
        return f2, pred_next, pred_action_logit


@REWARD_MODEL_REGISTRY.register('icm')
class ICMRewardModel(BaseRewardModel):

    config = dict(
        type='icm',
        intrinsic_reward_type='add',
        learning_rate=1e-3,
        obs_shape=4,
        action_shape=2,
        batch_size=64,
        hidden_size_list=[64, 64],
        update_per_collect=10,
        reverse_scale=1,
        intrinsic_reward_weight=0.003,
        extrinsic_reward_norm=True,
        extrinsic_reward_norm_max=1,
    )

    def __init__(self, config: EasyDict, device: str = 'cpu', tb_logger=None):
        super().__init__()
        self.cfg = EasyDict(copy.deepcopy(self.config))
        self.cfg.update(config or {})
        self.device = device
        self.reward_model = IcmNetwork(self.cfg.obs_shape, self.cfg.hidden_size_list, self.cfg.action_shape).to(device)
        self.opt = torch.optim.Adam(self.reward_model.parameters(), lr=self.cfg.learning_rate)
        self.train_data = []

    def collect_data(self, data: list) -> None:
        self.train_data.extend(
            [{'obs': d['obs'], 'next_obs': d['next_obs'], 'action': d['action']} for d in data]
        )

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
            a_onehot = F.one_hot(action, self.cfg.action_shape).float()
            real_next, pred_next, pred_logit = self.reward_model(obs, next_obs, a_onehot)
            fwd_loss = F.mse_loss(pred_next, real_next.detach())
            inv_loss = F.cross_entropy(pred_logit, action)
            loss = fwd_loss + self.cfg.reverse_scale * inv_loss
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()

    def estimate(self, data: list) -> List[dict]:
        out = self.reward_deepcopy(data)
        obs = torch.stack([torch.as_tensor(d['obs'], dtype=torch.float32) for d in out]).to(self.device)
        next_obs = torch.stack([torch.as_tensor(d['next_obs'], dtype=torch.float32) for d in out]).to(self.device)
        action = torch.stack([torch.as_tensor(d['action']).reshape(()) for d in out]).long().to(self.device)
        a_onehot = F.one_hot(action, self.cfg.action_shape).float()
        with torch.no_grad():
            real_next, pred_next, _ = self.reward_model(obs, next_obs, a_onehot)
            icm_reward = (real_next - pred_next).pow(2).mean(1)
            icm_reward = (icm_reward - icm_reward.min()) / (icm_reward.max() - icm_reward.min() + 1e-8)
        for item, ir in zip(out, icm_reward):
            rew = item['reward']
            if self.cfg.extrinsic_reward_norm:
                rew = rew / self.cfg.extrinsic_reward_norm_max
            ir = ir.item() * self.cfg.intrinsic_reward_weight
            if self.cfg.intrinsic_reward_type == 'add':
                item['reward'] = rew + ir
            elif self.cfg.intrinsic_reward_type == 'new':
                item['intrinsic_reward'] = torch.as_tensor([ir])
            else:
                item['reward'] = torch.as_tensor([ir])
        return out
