"""Adapt any gym-style env object into a BaseEnv, applying a cfg-driven
wrapper stack.

Parity: reference ding/envs/env/ding_env_wrapper.py:17 (DingEnvWrapper).
"""
import copy
from typing import Any, List, Optional, Union

import numpy as np

from ding.utils import ENV_WRAPPER_REGISTRY, EasyDict
from ding.torch_utils import to_ndarray
from .base_env import BaseEnv, BaseEnvTimestep
from ..env_wrappers.env_wrappers import EvalEpisodeReturnWrapper


class DingEnvWrapper(BaseEnv):

    def __init__(self, env: Any = None, cfg: dict = None, seed_api: bool = True, caller: str = 'collector'):
        self._cfg = EasyDict(cfg or {})
        self._seed_api = seed_api
        self._caller = caller
        if 'act_scale' not in self._cfg:
            self._cfg.act_scale = False
        self._env_fn = None
        self._raw_env = env
        self._env = None
        self._seed = None
        self._dynamic_seed = True
        self._init_flag = False
        self._observation_space = None
        self._action_space = None
        self._reward_space = None
        if env is not None:
            self._init_env(env)

    def _init_env(self, env):
        self._env = env
        wrapper_cfgs = self._cfg.get('env_wrapper', [])
        if isinstance(wrapper_cfgs, str):
            wrapper_cfgs = []
        for w in wrapper_cfgs:
            if isinstance(w, dict):
                cls = ENV_WRAPPER_REGISTRY.get(w['type'])
                self._env = cls(self._env, **w.get('kwargs', {}))
            else:
                self._env = w(self._env)
        if not any(isinstance(x, EvalEpisodeReturnWrapper) for x in self._iter_wrappers()):
            self._env = EvalEpisodeReturnWrapper(self._env)
        self._observation_space = getattr(self._env, 'observation_space', None)
        self._action_space = getattr(self._env, 'action_space', None)
        self._reward_space = getattr(self._env, 'reward_space', None)
        self._init_flag = True

    def _iter_wrappers(self):
        env = self._env
        while hasattr(env, 'env'):
            yield env
            env = env.env

    def reset(self) -> np.ndarray:
        if not self._init_flag:
            self._init_env(self._raw_env if self._raw_env is not None else self._env_fn())
        if self._seed is not None:
            seed = self._seed + np.random.randint(0, 100) if self._dynamic_seed else self._seed
            if self._seed_api and hasattr(self._env, 'seed'):
                try:
                    self._env.seed(seed)
                except TypeError:
                    pass
        obs = self._env.reset()
        return to_ndarray(obs)

    def step(self, action: Any) -> BaseEnvTimestep:
        action = self._judge_action_type(action)
        if self._cfg.act_scale:
            low, high = self._action_space.low, self._action_space.high
            action = low + (np.tanh(action) + 1) / 2 * (high - low)
        obs, rew, done, info = self._env.step(action)
        obs = to_ndarray(obs)
        rew = to_ndarray([rew], dtype=np.float32)
        return BaseEnvTimestep(obs, rew, done, info)

    def _judge_action_type(self, action):
        import torch
        if isinstance(action, torch.Tensor):
            action = action.cpu().numpy()
        if isinstance(action, np.ndarray) and action.shape == (1, ) and \
                getattr(self._action_space, 'n', None) is not None:
            action = int(action[0])
        elif isinstance(action, np.ndarray) and action.ndim == 0:
            action = action.item()
        return action

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed
        np.random.seed(seed)

    def close(self) -> None:
        if self._init_flag and hasattr(self._env, 'close'):
            self._env.close()
        self._init_flag = False

    def random_action(self) -> np.ndarray:
        a = self._action_space.sample()
        if isinstance(a, (int, np.integer)):
            a = np.array([a], dtype=np.int64)
        return a

    @property
    def observation_space(self):
        return self._observation_space

    @property
    def action_space(self):
        return self._action_space

    @property
    def reward_space(self):
        return self._reward_space

    def clone(self, caller: str = 'collector') -> 'DingEnvWrapper':
        return copy.deepcopy(self)

    def __repr__(self) -> str:
        return "DingEnvWrapper(" + type(self._raw_env).__name__ + ")"
