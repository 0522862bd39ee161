"""Bit-flip goal environment (the canonical HER benchmark): flip bits of an
n-bit state to match a random goal; reward only on exact match. Obs is the
concatenation [state, goal] so hindsight relabeling can substitute achieved
goals.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('bitflip')
class BitFlipEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        self.n_bits = cfg.get('n_bits', 8)
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        self._observation_space = Box(0, 1, (2 * self.n_bits, ))
        self._action_space = Discrete(self.n_bits)
        self._reward_space = Box(0, 1, (1, ))

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._state = self._rng.randint(0, 2, self.n_bits).astype(np.float32)
        self._goal = self._rng.randint(0, 2, self.n_bits).astype(np.float32)
        while (self._state == self._goal).all():
            self._goal = self._rng.randint(0, 2, self.n_bits).astype(np.float32)
        self._t = 0
        self._return = 0.0
        return np.concatenate([self._state, self._goal])

    def step(self, action: Any) -> BaseEnvTimestep:
        a = int(np.asarray(action).item())
        self._state[a] = 1.0 - self._state[a]
        self._t += 1
        success = bool((self._state == self._goal).all())
        done = success or self._t >= self.n_bits * 2
        reward = 1.0 if success else 0.0
        self._return += reward
        info = {'eval_episode_return': self._return, 'success': success} if done else {}
        return BaseEnvTimestep(
            np.concatenate([self._state, self._goal]), np.array([reward], dtype=np.float32), done, info
        )

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
        return f"BitFlipEnv({self.n_bits} bits)"
