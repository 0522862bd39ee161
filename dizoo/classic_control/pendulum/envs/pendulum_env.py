"""Pendulum-v1 dynamics (continuous control smoke env), gym-equivalent:
torque in [-2, 2], obs (cos th, sin th, thdot), reward
-(th^2 + 0.1 thdot^2 + 0.001 u^2), 200-step episodes.
"""
import math
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box
from ding.utils import ENV_REGISTRY


def _angle_normalize(x):
    return ((x + np.pi) % (2 * np.pi)) - np.pi


@ENV_REGISTRY.register('pendulum')
class PendulumEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_speed = 8.0
        self._max_torque = 2.0
        self._dt = 0.05
        self._g = 10.0
        self._m = 1.0
        self._l = 1.0
        self._max_step = self._cfg.get('max_step', 200)
        self._act_scale = self._cfg.get('act_scale', True)
        self._observation_space = Box(
            low=np.array([-1.0, -1.0, -8.0], dtype=np.float32),
            high=np.array([1.0, 1.0, 8.0], dtype=np.float32), shape=(3, )
        )
        self._action_space = Box(-self._max_torque, self._max_torque, (1, ))
        self._reward_space = Box(-17.0, 0.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        high = np.array([np.pi, 1.0])
        self._state = self._rng.uniform(-high, high)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._get_obs()

    def _get_obs(self):
        th, thdot = self._state
        return np.array([math.cos(th), math.sin(th), thdot], dtype=np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        action = np.asarray(action, dtype=np.float32).reshape(-1)
        if self._act_scale:
            u = np.clip(action, -1, 1)[0] * self._max_torque
        else:
            u = np.clip(action, -self._max_torque, self._max_torque)[0]
        th, thdot = self._state
        cost = _angle_normalize(th) ** 2 + 0.1 * thdot ** 2 + 0.001 * u ** 2
        newthdot = thdot + (3 * self._g / (2 * self._l) * math.sin(th) + 3.0 / (self._m * self._l ** 2) * u) * self._dt
        newthdot = np.clip(newthdot, -self._max_speed, self._max_speed)
        newth = th + newthdot * self._dt
        self._state = np.array([newth, newthdot])
        self._step_count += 1
        done = self._step_count >= self._max_step
        reward = -float(cost)
        self._eval_episode_return += reward
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(self._get_obs(), np.array([reward], dtype=np.float32), done, info)

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return self._action_space.sample()

    @property
    def observation_space(self):
        return self._observation_space

    @property
    def action_space(self):
        return self._action_space

    @property
    def reward_space(self):
        return self._reward_space

    def __repr__(self) -> str:
        return "PendulumEnv"
