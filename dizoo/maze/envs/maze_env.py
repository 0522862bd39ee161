"""Maze env for procedure cloning (reference dizoo/maze/envs/maze_env.py):
a seeded NxN wall maze; obs is the reference's 8-channel stacked map
[agent one-hot, target one-hot, walls, empty, + 4 action-history planes
kept zero here]; actions up/down/left/right; reward 1 only on reaching the
target (stop_value 1). Exposes ``maze`` (walls grid) and
``target_location`` so ding.utils.get_vi_sequence can run value-iteration
BFS over it (the PC-BFS training signal).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('maze')
class MazeEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.size = int(self._cfg.get('size', 16))
        self._max_step = self._cfg.get('max_step', 4 * self.size * self.size)
        self._observation_space = Box(0.0, 1.0, (8, self.size, self.size))
        self._action_space = Discrete(4)
        self._reward_space = Box(0.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        self.maze = np.zeros((self.size, self.size), dtype=np.int64)
        self.target_location = (self.size - 1, self.size - 1)

    def _generate(self, rng) -> None:
        """Recursive-backtracker maze on the odd lattice (walls = 1)."""
        n = self.size
        self.maze = np.ones((n, n), dtype=np.int64)
        start = (0, 0)
        self.maze[start] = 0
        stack, seen = [start], {start}
        while stack:
            r, c = stack[-1]
            nbrs = [
                (r + dr, c + dc) for dr, dc in ((0, 2), (0, -2), (2, 0), (-2, 0))
                if 0 <= r + dr < n and 0 <= c + dc < n and (r + dr, c + dc) not in seen
            ]
            if not nbrs:
                stack.pop()
                continue
            nr, nc = nbrs[rng.randint(len(nbrs))]
            self.maze[(r + nr) // 2, (c + nc) // 2] = 0
            self.maze[nr, nc] = 0
            seen.add((nr, nc))
            stack.append((nr, nc))
        open_cells = list(zip(*np.nonzero(self.maze == 0)))
        self.target_location = tuple(open_cells[-1])
        self._agent = tuple(open_cells[0])

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._generate(self._rng)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        o = np.zeros((8, self.size, self.size), dtype=np.float32)
        o[0, self._agent[0], self._agent[1]] = 1.0
        o[1, self.target_location[0], self.target_location[1]] = 1.0
        o[2] = (self.maze == 1).astype(np.float32)
        o[3] = (self.maze == 0).astype(np.float32)
        return o

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.reshape(-1)[0])
        dr, dc = [(-1, 0), (1, 0), (0, -1), (0, 1)][int(action)]
        r, c = self._agent[0] + dr, self._agent[1] + dc
        if 0 <= r < self.size and 0 <= c < self.size and self.maze[r, c] == 0:
            self._agent = (r, c)
        self._step_count += 1
        done = False
        reward = 0.0
        if self._agent == tuple(self.target_location):
            reward, done = 1.0, True
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

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
        return f"MazeEnv({self.size})"
