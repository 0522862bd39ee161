"""Rocket landing/hovering (reference dizoo/rocket/envs wrapping
rocket-recycling). Native 2D rocket dynamics: thrust (throttle) + nozzle
gimbal torque; tasks 'hover' (stay near a target altitude) and 'landing'
(touch down slowly on the pad). Obs 8 [x, y, vx, vy, theta, omega,
fuel, task-target-alt]; Discrete(9) = 3 throttle x 3 gimbal.
"""
from typing import Any
import math

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('rocket')
class RocketEnv(BaseEnv):

    G = -0.8
    THRUSTS = [0.0, 1.0, 2.0]
    GIMBALS = [-1.0, 0.0, 1.0]

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.task = self._cfg.get('task', 'hover')
        self._max_step = self._cfg.get('max_step', 500)
        self._observation_space = Box(-np.inf, np.inf, (8, ))
        self._action_space = Discrete(9)
        self._reward_space = Box(-1.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self.x = self._rng.uniform(-0.3, 0.3)
        self.y = 1.5 if self.task == 'landing' else self._rng.uniform(0.8, 1.2)
        self.vx, self.vy = 0.0, self._rng.uniform(-0.2, 0.0)
        self.theta = self._rng.uniform(-0.15, 0.15)
        self.omega = 0.0
        self.fuel = 1.0
        self.target_alt = 1.0 if self.task == 'hover' else 0.0
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        return np.array([
            self.x, self.y, self.vx, self.vy, self.theta, self.omega, self.fuel, self.target_alt
        ], dtype=np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        if hasattr(action, 'reshape'):
            action = int(np.asarray(action).reshape(-1)[0])
        throttle = self.THRUSTS[action // 3] * (1.0 if self.fuel > 0 else 0.0)
        gimbal = self.GIMBALS[action % 3]
        dt = 0.05
        self.fuel = max(0.0, self.fuel - 0.002 * throttle)
        ax = -math.sin(self.theta) * throttle
        ay = math.cos(self.theta) * throttle + self.G
        self.omega += gimbal * 2.0 * dt - 0.3 * self.omega * dt
        self.theta += self.omega * dt
        self.vx += ax * dt
        self.vy += ay * dt
        self.x += self.vx * dt
        self.y += self.vy * dt
        self._step_count += 1

        done = False
        if self.task == 'hover':
            err = abs(self.y - self.target_alt) + 0.5 * abs(self.x) + 0.3 * abs(self.theta)
            reward = max(0.0, 1.0 - err) * 0.02
            if self.y <= 0.0 or self.y > 3.0 or abs(self.theta) > 1.0:
                reward = -1.0
                done = True
        else:  # landing
            reward = -0.001 * (abs(self.vx) + abs(self.theta))
            if self.y <= 0.0:
                soft = abs(self.vy) < 0.3 and abs(self.vx) < 0.2 and abs(self.theta) < 0.2 and abs(self.x) < 0.3
                reward = 1.0 if soft else -1.0
                done = True
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

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
        return f"RocketEnv({self.task})"
