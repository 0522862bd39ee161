"""bsuite-style memory-length env: a cue shown at t=0 must be reproduced at
t=T-1; all intermediate observations are noise. Pure recurrence test for
R2D2/NGU/GTrXL (reward only depends on remembering the first step).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('memory_len')
class MemoryLenEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        self.horizon = cfg.get('memory_length', 8)
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        # obs: [is_first, countdown/T, cue (+/-1 at t=0 else 0)]
        self._observation_space = Box(-1, 1, (3, ))
        self._action_space = Discrete(2)
        self._reward_space = Box(-1, 1, (1, ))

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._cue = int(self._rng.randint(0, 2))
        self._t = 0
        return np.array([1.0, 1.0, 1.0 if self._cue else -1.0], dtype=np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        self._t += 1
        done = self._t >= self.horizon
        reward = 0.0
        if done:
            a = int(np.asarray(action).item())
            reward = 1.0 if a == self._cue else -1.0
        obs = np.array([0.0, 1.0 - self._t / self.horizon, 0.0], dtype=np.float32)
        info = {'eval_episode_return': reward} if done else {}
        return BaseEnvTimestep(obs, np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

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
        return f"MemoryLenEnv(T={self.horizon})"
