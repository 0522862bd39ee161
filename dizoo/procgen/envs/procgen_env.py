"""Procgen (coinrun / maze / bigfish) with native procedural generation.

The procgen binaries are unavailable offline; this env keeps the reference
interface (dizoo/procgen/envs/procgen_env.py: obs [3, 64, 64] uint8, 15
discrete actions, per-level procedural generation driven by an integer
level seed with ``start_level``/``num_levels``) and implements three
game-alikes on a hidden 8x8 grid rendered to 64x64 RGB:

* coinrun: reach the coin past level-seeded walls; +10 on the coin (episode
  ends), sparse otherwise. stop_value 10.
* maze: DFS-generated perfect maze, +10 at the goal.
* bigfish: eat smaller dots (+1 each), episode ends if a bigger one hits
  you; stop_value 40.

Level identity is fully determined by the level seed, so PLR's level-replay
sampling (ding/data/level_replay) has real generalization structure to
exploit: walls/maze/fish layouts differ per level.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

CELL = 8  # render scale: 8x8 grid -> 64x64 image


@ENV_REGISTRY.register('procgen')
class ProcgenEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._game = self._cfg.get('env_id', 'coinrun')
        self._start_level = int(self._cfg.get('start_level', 0))
        self._num_levels = int(self._cfg.get('num_levels', 0))  # 0 = unbounded
        self._max_step = self._cfg.get('max_step', 200)
        self._observation_space = Box(0.0, 1.0, (3, 64, 64), dtype=np.float32)
        self._action_space = Discrete(15)
        self._reward_space = Box(-1, 10, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        self._forced_level = None

    # procgen action ids: subset used — 1 left, 7 right, 5 down, 11 up (the
    # rest are no-ops here, as many procgen actions alias per game)
    _MOVES = {1: (0, -1), 7: (0, 1), 5: (1, 0), 11: (-1, 0)}

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reseed(self, level: int) -> None:
        """PLR hook: force the next reset onto a specific level seed."""
        self._forced_level = int(level)

    def _gen_level(self, level: int) -> None:
        rng = np.random.RandomState(level)
        self._walls = np.zeros((8, 8), dtype=bool)
        if self._game == 'maze':
            # DFS perfect maze on the 4x4 coarse grid expanded to 8x8
            self._walls[:] = True
            stack = [(0, 0)]
            seen = {(0, 0)}
            self._walls[0, 0] = False
            while stack:
                r, c = stack[-1]
                nbrs = [(r + dr, c + dc) for dr, dc in ((0, 2), (0, -2), (2, 0), (-2, 0))
                        if 0 <= r + dr < 8 and 0 <= c + dc < 8 and (r + dr, c + dc) not in seen]
                if not nbrs:
                    stack.pop()
                    continue
                nr, nc = nbrs[rng.randint(len(nbrs))]
                self._walls[(r + nr) // 2, (c + nc) // 2] = False
                self._walls[nr, nc] = False
                seen.add((nr, nc))
                stack.append((nr, nc))
            self._agent = [0, 0]
            self._goal = [7 - (7 % 2), 7 - (7 % 2)]  # (6, 6) is open by construction
        elif self._game == 'bigfish':
            self._agent = [4, 0]
            self._size = 1
            self._fish = []  # (row, col, size)
            for _ in range(6):
                self._fish.append([rng.randint(0, 8), rng.randint(2, 8), rng.randint(1, 4)])
            self._goal = None
        else:  # coinrun
            for _ in range(6):
                r, c = rng.randint(0, 8), rng.randint(1, 7)
                self._walls[r, c] = True
            self._agent = [rng.randint(0, 8), 0]
            self._goal = [rng.randint(0, 8), 7]
            self._walls[self._agent[0], 0] = False
            self._walls[self._goal[0], 7] = False

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        if self._forced_level is not None:
            level = self._forced_level
            self._forced_level = None
        elif self._num_levels > 0:
            level = self._start_level + self._rng.randint(0, self._num_levels)
        else:
            level = self._rng.randint(0, 2 ** 31 - 1)
        self._level = level
        self._gen_level(level)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._render()

    def _render(self) -> np.ndarray:
        img = np.zeros((3, 64, 64), dtype=np.uint8)
        img[2] = 40  # background
        for r in range(8):
            for c in range(8):
                if self._walls[r, c]:
                    img[:, r * CELL:(r + 1) * CELL, c * CELL:(c + 1) * CELL] = 100
        if self._game == 'bigfish':
            for r, c, s in self._fish:
                ch = 0 if s > self._size else 1
                img[ch, r * CELL:(r + 1) * CELL, c * CELL:(c + 1) * CELL] = min(150 + 13 * s, 255)
        elif self._goal is not None:
            gr, gc = self._goal
            img[1, gr * CELL:(gr + 1) * CELL, gc * CELL:(gc + 1) * CELL] = 255  # coin/goal green
        ar, ac = self._agent
        img[0, ar * CELL:(ar + 1) * CELL, ac * CELL:(ac + 1) * CELL] = 255  # agent red
        # reference env emits float32 frames (procgen_env.py to_ndarray(...float32))
        return img.astype(np.float32) / 255.0

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.reshape(-1)[0])
        action = int(action)
        dr, dc = self._MOVES.get(action, (0, 0))
        nr, nc = self._agent[0] + dr, self._agent[1] + dc
        if 0 <= nr < 8 and 0 <= nc < 8 and not self._walls[nr, nc]:
            self._agent = [nr, nc]
        reward = 0.0
        done = False
        if self._game == 'bigfish':
            for f in list(self._fish):
                if self._rng.rand() < 0.3:  # fish drift
                    f[0] = int(np.clip(f[0] + self._rng.randint(-1, 2), 0, 7))
                    f[1] = int(np.clip(f[1] + self._rng.randint(-1, 2), 0, 7))
                if f[0] == self._agent[0] and f[1] == self._agent[1]:
                    if f[2] <= self._size:
                        reward += 1.0
                        self._size = min(self._size + 1, 8)
                        self._fish.remove(f)
                        self._fish.append([self._rng.randint(0, 8), self._rng.randint(0, 8),
                                           self._rng.randint(1, 5)])
                    else:
                        done = True
            if not self._fish:
                done = True
        elif self._agent == self._goal:
            reward = 10.0
            done = True
        self._step_count += 1
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {'level': self._level}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(self._render(), np.array([reward], dtype=np.float32), done, info)

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
        return f"ProcgenEnv({self._game})"
