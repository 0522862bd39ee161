"""MetaDrive-style driving ("metadrive-lite", reference dizoo/metadrive
wrapping the MetaDrive simulator). A seeded curvy road; the car controls
[steering, throttle] in [-1, 1]^2; obs = ego state + 12-beam lane lidar;
reward = forward speed along the lane - lateral/heading penalties; episode
ends off-road or at route end (macro PPO lane of the reference).
"""
from typing import Any
import math

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('metadrive')
class MetaDriveLiteEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_step = self._cfg.get('max_step', 500)
        self.route_len = 100.0
        self.lane_width = 1.5
        self.n_beams = 12
        self._observation_space = Box(-np.inf, np.inf, (5 + self.n_beams, ))
        self._action_space = Box(-1.0, 1.0, (2, ))
        self._reward_space = Box(-5.0, 2.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def _lane_heading(self, s: float) -> float:
        """Road centerline heading at arclength s (seeded curvature)."""
        return float(sum(a * math.sin(w * s + p) for a, w, p in self._curves))

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        self._curves = [(self._rng.uniform(0.1, 0.3), self._rng.uniform(0.05, 0.15),
                         self._rng.uniform(0, 6.28)) for _ in range(3)]
        self.s = 0.0        # progress along lane
        self.d = 0.0        # lateral offset
        self.heading_err = 0.0
        self.speed = 0.0
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        # lidar: distance to lane border along beams (analytic for straight
        # border approximation at current offset)
        beams = np.zeros(self.n_beams, dtype=np.float32)
        for i in range(self.n_beams):
            ang = -math.pi / 2 + i * math.pi / (self.n_beams - 1) + self.heading_err
            sin = math.sin(ang)
            if sin > 1e-3:
                beams[i] = min((self.lane_width - self.d) / sin, 10.0)
            elif sin < -1e-3:
                beams[i] = min((self.lane_width + self.d) / -sin, 10.0)
            else:
                beams[i] = 10.0
        state = np.array([
            self.speed / 5.0, self.d / self.lane_width, self.heading_err,
            self._lane_heading(self.s + 2) - self._lane_heading(self.s),  # upcoming curvature
            self.s / self.route_len
        ], dtype=np.float32)
        return np.concatenate([state, beams / 10.0])

    def step(self, action: Any) -> BaseEnvTimestep:
        a = np.clip(np.asarray(action, dtype=np.float64).reshape(-1), -1, 1)
        steer, throttle = float(a[0]), float(a[1])
        dt = 0.1
        self.speed = float(np.clip(self.speed + throttle * 2.0 * dt - 0.1 * self.speed * dt, 0.0, 5.0))
        self.heading_err += steer * 0.8 * dt * (1 + self.speed / 5)
        lane_turn = self._lane_heading(self.s + self.speed * dt) - self._lane_heading(self.s)
        self.heading_err -= lane_turn
        self.d += self.speed * math.sin(self.heading_err) * dt
        self.s += self.speed * math.cos(self.heading_err) * dt
        self._step_count += 1
        reward = 0.1 * self.speed * math.cos(self.heading_err) \
            - 0.1 * abs(self.d) - 0.05 * abs(self.heading_err)
        done = False
        if abs(self.d) > self.lane_width:
            reward = -5.0
            done = True
        if self.s >= self.route_len:
            reward += 2.0
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
        return "MetaDriveLiteEnv"
