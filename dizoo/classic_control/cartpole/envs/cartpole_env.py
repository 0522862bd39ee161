"""CartPole-v0 physics implemented from the classic OpenAI/Barto-Sutton
cart-pole equations (no gym offline). Matches gym's CartPole-v0 dynamics:
force +-10, dt 0.02, Euler integration, termination at |x|>2.4 or
|theta|>12 deg, max 200 steps, reward 1 per step.

Parity target: dizoo/classic_control/cartpole/envs/cartpole_env.py in the
reference (stop_value 195 per cartpole_dqn_config.py:9).
"""
import math
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('cartpole')
class CartPoleEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._gravity = 9.8
        self._masscart = 1.0
        self._masspole = 0.1
        self._total_mass = self._masscart + self._masspole
        self._length = 0.5
        self._polemass_length = self._masspole * self._length
        self._force_mag = 10.0
        self._tau = 0.02
        self._theta_threshold = 12 * 2 * math.pi / 360
        self._x_threshold = 2.4
        self._max_step = self._cfg.get('max_step', 200)
        self._observation_space = Box(
            low=np.array([-4.8, -np.inf, -0.42, -np.inf], dtype=np.float32),
            high=np.array([4.8, np.inf, 0.42, np.inf], dtype=np.float32),
            shape=(4, ),
            dtype=np.float32,
        )
        self._action_space = Discrete(2)
        self._reward_space = Box(0.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        self._init_flag = False

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        self._state = self._rng.uniform(-0.05, 0.05, size=(4, ))
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._init_flag = True
        return self._state.astype(np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.item())
        action = int(action)
        x, x_dot, theta, theta_dot = self._state
        force = self._force_mag if action == 1 else -self._force_mag
        costheta, sintheta = math.cos(theta), math.sin(theta)
        temp = (force + self._polemass_length * theta_dot ** 2 * sintheta) / self._total_mass
        thetaacc = (self._gravity * sintheta - costheta * temp) / (
            self._length * (4.0 / 3.0 - self._masspole * costheta ** 2 / self._total_mass)
        )
        xacc = temp - self._polemass_length * thetaacc * costheta / self._total_mass
        x = x + self._tau * x_dot
        x_dot = x_dot + self._tau * xacc
        theta = theta + self._tau * theta_dot
        theta_dot = theta_dot + self._tau * thetaacc
        self._state = np.array([x, x_dot, theta, theta_dot])
        self._step_count += 1
        done = bool(
            x < -self._x_threshold or x > self._x_threshold or theta < -self._theta_threshold
            or theta > self._theta_threshold or self._step_count >= self._max_step
        )
        reward = 1.0
        self._eval_episode_return += reward
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(
            self._state.astype(np.float32), np.array([reward], dtype=np.float32), done, info
        )

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def close(self) -> None:
        self._init_flag = False

    def random_action(self) -> np.ndarray:
        return np.array([self._action_space.sample()], dtype=np.int64)

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
        return "CartPoleEnv"
