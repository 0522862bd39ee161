"""Quadrotor hover (reference dizoo/gym_pybullet_drones wrapping
gym-pybullet-drones takeoff/hover). Native point-mass quadrotor with
per-rotor thrust mixing: obs 12 [pos(3), rpy(3), vel(3), ang vel(3)],
Box(4) normalized rotor thrusts, reward = 1 - distance to the hover point
with crash termination.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('drone_hover')
class DroneHoverEnv(BaseEnv):

    G = 9.8

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_step = self._cfg.get('max_step', 400)
        self.target = np.array([0.0, 0.0, 1.0])
        self._observation_space = Box(-np.inf, np.inf, (12, ))
        self._action_space = Box(-1.0, 1.0, (4, ))
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
            self._action_space.seed(seed)
        self.pos = np.array([0.0, 0.0, 1.0]) + self._rng.uniform(-0.2, 0.2, 3)
        self.rpy = self._rng.uniform(-0.1, 0.1, 3)
        self.vel = np.zeros(3)
        self.ang = np.zeros(3)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        return np.concatenate([self.pos, self.rpy, self.vel, self.ang]).astype(np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        a = np.clip(np.asarray(action, dtype=np.float64).reshape(-1), -1, 1)
        dt = 1.0 / 48.0
        # rotor mixing: total thrust + roll/pitch/yaw moments
        thrust = self.G + 4.0 * a.mean()
        roll_m = (a[1] + a[2] - a[0] - a[3]) * 4.0
        pitch_m = (a[0] + a[1] - a[2] - a[3]) * 4.0
        yaw_m = (a[0] + a[2] - a[1] - a[3]) * 1.0
        self.ang += np.array([roll_m, pitch_m, yaw_m]) * dt - 0.5 * self.ang * dt
        self.rpy += self.ang * dt
        # body-z thrust in world frame (small-angle)
        acc = np.array([
            thrust * self.rpy[1], -thrust * self.rpy[0], thrust - self.G
        ])
        self.vel += acc * dt - 0.1 * self.vel * dt
        self.pos += self.vel * dt
        self._step_count += 1
        d = float(np.linalg.norm(self.pos - self.target))
        reward = max(0.0, 1.0 - d) * 0.05
        done = False
        if self.pos[2] <= 0.0 or d > 2.0 or np.abs(self.rpy[:2]).max() > 1.0:
            reward = -1.0
            done = True
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return self._action_space.sample().astype(np.float32)

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
        return "DroneHoverEnv"
