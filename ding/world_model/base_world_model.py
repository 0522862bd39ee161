"""World-model ABCs for model-based RL.

Parity: reference ding/world_model/base_world_model.py (WorldModel:27,
DynaWorldModel:141, DreamWorldModel:269, HybridWorldModel:351).
"""
import copy
from abc import ABC, abstractmethod
from typing import Any, Callable, Optional, Tuple, Union

import numpy as np
import torch

from ding.utils import WORLD_MODEL_REGISTRY, EasyDict, deep_merge_dicts, import_module
from ding.worker.replay_buffer.naive_buffer import NaiveReplayBuffer


def get_rollout_length_scheduler(cfg: EasyDict) -> Callable[[int], int]:
    """Linear rollout-length schedule (MBPO-style)."""
    if cfg is None:
        return lambda step: 1
    x0, x1 = cfg.rollout_start_step, cfg.rollout_end_step
    y0, y1 = cfg.rollout_length_min, cfg.rollout_length_max

    def scheduler(envstep: int) -> int:
        if envstep <= x0:
            return y0
        if envstep >= x1:
            return y1
        return int(y0 + (y1 - y0) * (envstep - x0) / (x1 - x0))

    return scheduler


class WorldModel(ABC):
    """train/eval on env data; step() produces imagined transitions."""

    config = dict(
        train_freq=250,
        eval_freq=250,
        cuda=False,
        rollout_length_scheduler=dict(
            type='linear',
            rollout_start_step=20000,
            rollout_end_step=150000,
            rollout_length_min=1,
            rollout_length_max=25,
        ),
    )

    def __init__(self, cfg: EasyDict, env=None, tb_logger=None):
        self.cfg = deep_merge_dicts(EasyDict(copy.deepcopy(self.config)), cfg or EasyDict({}))
        self.env = env
        self.tb_logger = tb_logger
        self._cuda = self.cfg.cuda and torch.cuda.is_available()
        self.device = 'cuda' if self._cuda else 'cpu'
        self.last_train_step = -1
        self.last_eval_step = -1
        self.rollout_length_scheduler = get_rollout_length_scheduler(self.cfg.get('rollout_length_scheduler'))

    @classmethod
    def default_config(cls) -> EasyDict:
        base = {}
        for klass in reversed(cls.__mro__):
            if hasattr(klass, 'config'):
                base = deep_merge_dicts(base, klass.config)
        return EasyDict(copy.deepcopy(base))

    def should_train(self, envstep: int) -> bool:
        if envstep - self.last_train_step >= self.cfg.train_freq:
            return True
        return False

    def should_eval(self, envstep: int) -> bool:
        return envstep - self.last_eval_step >= self.cfg.eval_freq and self.last_train_step >= 0

    @abstractmethod
    def train(self, env_buffer, envstep: int, train_iter: int) -> None:
        raise NotImplementedError

    @abstractmethod
    def eval(self, env_buffer, envstep: int, train_iter: int) -> None:
        raise NotImplementedError

    @abstractmethod
    def step(self, obs: torch.Tensor, action: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """(reward, next_obs, done) for a batch of imagined transitions."""
        raise NotImplementedError


class DynaWorldModel(WorldModel, ABC):
    """Dyna-style: fill an imagination buffer with model rollouts
    (reference base_world_model.py:141 fill_img_buffer)."""

    config = dict(
        other=dict(
            real_ratio=0.05,
            rollout_retain=4,
            rollout_batch_size=100000,
            imagination_buffer=dict(replay_buffer_size=600000, ),
        ),
    )

    def sample(self, env_buffer, img_buffer, batch_size: int, train_iter: int) -> list:
        """Mix real and imagined samples by real_ratio."""
        env_batch = int(batch_size * self.cfg.other.real_ratio)
        img_batch = batch_size - env_batch
        env_data = env_buffer.sample(env_batch, train_iter) or []
        img_data = img_buffer.sample(img_batch, train_iter) or []
        data = list(env_data) + list(img_data)
        if not data:
            return data
        # imagined transitions carry fewer meta keys than real ones; project
        # onto the common key set so collation stays uniform
        common = set(data[0].keys())
        for d in data[1:]:
            common &= set(d.keys())
        return [{k: d[k] for k in common} for d in data]

    def fill_img_buffer(self, policy, env_buffer, img_buffer, envstep: int, train_iter: int) -> None:
        rollout_length = self.rollout_length_scheduler(envstep)
        batch = env_buffer.sample(min(self.cfg.other.rollout_batch_size, env_buffer.count()), train_iter)
        if not batch:
            return
        obs = torch.stack([torch.as_tensor(d['obs'], dtype=torch.float32) for d in batch]).to(self.device)
        for _ in range(rollout_length):
            out = policy.forward({i: o for i, o in enumerate(obs)})
            action = torch.stack([out[i]['action'] for i in range(len(out))]).to(self.device)
            reward, next_obs, done = self.step(obs, action)
            for i in range(obs.shape[0]):
                img_buffer.push({
                    'obs': obs[i].cpu(),
                    'action': action[i].cpu(),
                    'reward': reward[i].reshape(-1).cpu(),
                    'next_obs': next_obs[i].cpu(),
                    'done': bool(done[i].item()),
                })
            keep = ~done.bool().reshape(-1)
            if keep.sum() == 0:
                break
            obs = next_obs[keep]


class DreamWorldModel(WorldModel, ABC):
    """Dream-style: differentiable rollout for gradient-through-dynamics
    (reference base_world_model.py:269 rollout)."""

    def rollout(self, obs: torch.Tensor, actor_fn: Callable, envstep: int,
                **kwargs) -> Tuple[torch.Tensor, ...]:
        """Differentiable imagination rollout for BPTT value gradients.

        Returns stacked tensors (reference base_world_model.py rollout):
        obss [N+1,B,O] (obss[0] real), actions [N+1,B,A] (incl. bootstrap
        action at the final state), rewards [N,B] (entropy-augmented),
        aug_rewards [N+1,B], dones [N,B]. Gradients flow to the POLICY only —
        the world model is frozen for the duration of the rollout."""
        import torch.nn as nn
        horizon = self.rollout_length_scheduler(envstep)
        if isinstance(self, nn.Module):
            self.requires_grad_(False)
        obss, actions, rewards, aug_rewards, dones = [obs], [], [], [], []
        for _ in range(horizon):
            action, aug_reward = actor_fn(obs)
            reward, obs, done = self.step(obs, action, **kwargs)
            reward = reward + aug_reward
            obss.append(obs)
            actions.append(action)
            rewards.append(reward)
            aug_rewards.append(aug_reward)
            dones.append(done)
        action, aug_reward = actor_fn(obs)
        actions.append(action)
        aug_rewards.append(aug_reward)
        if isinstance(self, nn.Module):
            self.requires_grad_(True)
        return (
            torch.stack(obss), torch.stack(actions), torch.stack(rewards),
            torch.stack(aug_rewards), torch.stack(dones)
        )


class HybridWorldModel(DreamWorldModel, DynaWorldModel, ABC):
    pass


def get_world_model_cls(cfg: EasyDict) -> type:
    """Registry lookup without construction (reference base_world_model.py:17)."""
    from ding.utils import import_module
    import_module(cfg.get('import_names', []))
    return WORLD_MODEL_REGISTRY.get(cfg.type)


def create_world_model(cfg: EasyDict, env=None, tb_logger=None) -> WorldModel:
    cfg = EasyDict(copy.deepcopy(cfg))
    if 'import_names' in cfg:
        import_module(cfg.import_names)
    return WORLD_MODEL_REGISTRY.build(cfg.type, cfg=cfg, env=env, tb_logger=tb_logger)
