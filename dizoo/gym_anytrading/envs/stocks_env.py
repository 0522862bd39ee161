"""gym-anytrading 'stocks-v0' implemented natively (reference
dizoo/gym_anytrading/envs/stocks_env.py). Prices are a seeded geometric
random walk with regime drift (no CSV data offline); the observation is the
window of (normalized price diff, position flag) pairs; actions
{0: sell/short, 1: buy/long}; reward = position * price change - commission
on flips; eval return is total profit.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('stocks')
class StocksEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.window = int(self._cfg.get('window_size', 20))
        self.horizon = int(self._cfg.get('eps_length', 200))
        self.commission = float(self._cfg.get('commission', 0.001))
        self._observation_space = Box(-np.inf, np.inf, (self.window * 2, ))
        self._action_space = Discrete(2)
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
        n = self.horizon + self.window + 1
        drift = np.repeat(self._rng.uniform(-0.001, 0.001, size=n // 50 + 1), 50)[:n]
        steps = drift + self._rng.randn(n) * 0.01
        self.prices = 100.0 * np.exp(np.cumsum(steps)).astype(np.float64)
        self.t = self.window
        self.position = 0  # -1 short, +1 long (0 only before first trade)
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        diffs = np.diff(self.prices[self.t - self.window:self.t + 1]) / self.prices[self.t - self.window:self.t]
        pos = np.full(self.window, float(self.position), dtype=np.float32)
        return np.stack([diffs.astype(np.float32) * 100.0, pos], axis=-1).reshape(-1)

    def step(self, action: Any) -> BaseEnvTimestep:
        if hasattr(action, 'reshape'):
            action = int(np.asarray(action).reshape(-1)[0])
        new_pos = 1 if int(action) == 1 else -1
        cost = self.commission if new_pos != self.position and self.position != 0 else 0.0
        self.position = new_pos
        prev = self.prices[self.t]
        self.t += 1
        ret = (self.prices[self.t] - prev) / prev
        reward = float(self.position * ret - cost)
        self._eval_episode_return += reward
        done = self.t >= self.window + self.horizon
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
        return "StocksEnv"
