"""FrozenLake-v1 implemented natively (gymnasium unavailable offline).

Exact 4x4 map and transition rules: actions LEFT/DOWN/RIGHT/UP, optional
slippery dynamics (uniform over the three non-opposite directions), reward 1
on reaching G, holes terminate with 0. Observation is the one-hot state
vector (16), matching the reference dizoo/frozen_lake/envs/frozen_lake_env.py
+ frozen_lake_dqn_config.py (obs_shape=16, action_shape=4).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

MAP_4x4 = ["SFFF", "FHFH", "FFFH", "HFFG"]


@ENV_REGISTRY.register('frozen_lake')
class FrozenLakeEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        desc = self._cfg.get('desc', None) or MAP_4x4
        self._desc = [list(row) for row in desc]
        self._nrow, self._ncol = len(self._desc), len(self._desc[0])
        self._n_states = self._nrow * self._ncol
        self._slippery = self._cfg.get('is_slippery', False)
        self._max_step = self._cfg.get('max_step', 100)
        self._observation_space = Box(0.0, 1.0, (self._n_states, ))
        self._action_space = Discrete(4)  # 0 left, 1 down, 2 right, 3 up
        self._reward_space = Box(0.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        self._s = 0
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._one_hot(self._s)

    def _one_hot(self, s: int) -> np.ndarray:
        v = np.zeros(self._n_states, dtype=np.float32)
        v[s] = 1.0
        return v

    def _move(self, s: int, a: int) -> int:
        row, col = divmod(s, self._ncol)
        if a == 0:
            col = max(col - 1, 0)
        elif a == 1:
            row = min(row + 1, self._nrow - 1)
        elif a == 2:
            col = min(col + 1, self._ncol - 1)
        elif a == 3:
            row = max(row - 1, 0)
        return row * self._ncol + col

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.item())
        action = int(action)
        if self._slippery:
            # gym semantics: actual direction uniform over {a-1, a, a+1}
            action = int(self._rng.choice([(action - 1) % 4, action, (action + 1) % 4]))
        self._s = self._move(self._s, action)
        row, col = divmod(self._s, self._ncol)
        tile = self._desc[row][col]
        self._step_count += 1
        reward = 1.0 if tile == 'G' else 0.0
        done = tile in ('G', 'H') or self._step_count >= self._max_step
        self._eval_episode_return += reward
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(self._one_hot(self._s), np.array([reward], dtype=np.float32), done, info)

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

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
        return "FrozenLakeEnv"
