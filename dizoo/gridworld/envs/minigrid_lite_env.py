"""Sparse-reward gridworld (minigrid-style, self-contained): an agent on an
N x N grid must reach a goal; reward only on success. The obs is an
egocentric one-hot map + agent direction, which makes it a standard
exploration benchmark for RND/ICM/NGU reward models without the external
minigrid dependency.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('minigrid_lite')
class MiniGridLiteEnv(BaseEnv):
    """Actions: 0 left-turn, 1 right-turn, 2 forward. Obs: flattened
    [grid one-hot (empty/wall/goal/agent), dir one-hot]."""

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        self.n = cfg.get('grid_size', 8)
        self.max_step = cfg.get('max_step', 4 * self.n * self.n)
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        obs_dim = self.n * self.n * 4 + 4
        self._observation_space = Box(0, 1, (obs_dim, ))
        self._action_space = Discrete(3)
        self._reward_space = Box(0, 1, (1, ))

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._grid = np.zeros((self.n, self.n), dtype=np.int64)
        # border walls
        self._grid[0, :] = self._grid[-1, :] = self._grid[:, 0] = self._grid[:, -1] = 1
        # a random inner wall with a gap
        col = self._rng.randint(2, self.n - 2)
        gap = self._rng.randint(1, self.n - 1)
        self._grid[:, col] = 1
        self._grid[gap, col] = 0
        self._goal = (self.n - 2, self.n - 2)
        self._grid[self._goal] = 2
        self._pos = np.array([1, 1])
        self._dir = 0  # 0 E, 1 S, 2 W, 3 N
        self._step_count = 0
        self._return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        onehot = np.zeros((self.n, self.n, 4), dtype=np.float32)
        for v in range(3):
            onehot[..., v][self._grid == v] = 1.0
        onehot[self._pos[0], self._pos[1], 3] = 1.0
        d = np.zeros(4, dtype=np.float32)
        d[self._dir] = 1.0
        return np.concatenate([onehot.reshape(-1), d])

    def step(self, action: Any) -> BaseEnvTimestep:
        a = int(np.asarray(action).item())
        if a == 0:
            self._dir = (self._dir - 1) % 4
        elif a == 1:
            self._dir = (self._dir + 1) % 4
        else:
            delta = [(0, 1), (1, 0), (0, -1), (-1, 0)][self._dir]
            nxt = self._pos + delta
            if self._grid[nxt[0], nxt[1]] != 1:
                self._pos = nxt
        self._step_count += 1
        done = False
        reward = 0.0
        if tuple(self._pos) == self._goal:
            reward = 1.0 - 0.9 * self._step_count / self.max_step
            done = True
        elif self._step_count >= self.max_step:
            done = True
        self._return += reward
        info = {'eval_episode_return': self._return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

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
        return f"MiniGridLiteEnv({self.n}x{self.n})"
