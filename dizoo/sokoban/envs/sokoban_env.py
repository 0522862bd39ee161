"""Sokoban implemented natively (reference dizoo/sokoban/envs/sokoban_env.py
wraps gym-sokoban). A seeded room with B boxes and targets; actions
up/down/left/right push boxes; rewards: -0.1/step, +1 box-on-target,
-1 box-off-target, +10 all solved. Obs: 4-channel map [walls, targets,
boxes, player] at room_size x room_size.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('sokoban')
class SokobanEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.n = int(self._cfg.get('room_size', 7))
        self.num_boxes = int(self._cfg.get('num_boxes', 2))
        self._max_step = self._cfg.get('max_step', 120)
        self._observation_space = Box(0.0, 1.0, (4, self.n, self.n))
        self._action_space = Discrete(4)
        self._reward_space = Box(-1.0, 10.0, (1, ))
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
        n = self.n
        self.walls = np.zeros((n, n), dtype=bool)
        self.walls[0, :] = self.walls[-1, :] = self.walls[:, 0] = self.walls[:, -1] = True
        inner = [(r, c) for r in range(2, n - 2) for c in range(2, n - 2)]
        self._rng.shuffle(inner)
        # reverse-play generation: boxes start ON targets then get pulled off,
        # which guarantees solvability
        self.targets = set(inner[:self.num_boxes])
        self.boxes = set(self.targets)
        free = [p for p in inner[self.num_boxes:]]
        self.player = free[0] if free else (1, 1)
        for _ in range(8):  # pull moves
            boxes = list(self.boxes)
            b = boxes[self._rng.randint(len(boxes))]
            d = [(-1, 0), (1, 0), (0, -1), (0, 1)][self._rng.randint(4)]
            nb = (b[0] + d[0], b[1] + d[1])
            pp = (nb[0] + d[0], nb[1] + d[1])
            if not self._blocked(nb) and not self._blocked(pp) and nb not in self.boxes and pp not in self.boxes:
                self.boxes.remove(b)
                self.boxes.add(nb)
                self.player = pp
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._on_target = len(self.boxes & self.targets)
        return self._obs()

    def _blocked(self, p) -> bool:
        return not (0 <= p[0] < self.n and 0 <= p[1] < self.n) or bool(self.walls[p])

    def _obs(self) -> np.ndarray:
        o = np.zeros((4, self.n, self.n), dtype=np.float32)
        o[0][self.walls] = 1.0
        for t in self.targets:
            o[1][t] = 1.0
        for b in self.boxes:
            o[2][b] = 1.0
        o[3][self.player] = 1.0
        return o

    def step(self, action: Any) -> BaseEnvTimestep:
        if hasattr(action, 'reshape'):
            action = int(np.asarray(action).reshape(-1)[0])
        d = [(-1, 0), (1, 0), (0, -1), (0, 1)][int(action)]
        np_ = (self.player[0] + d[0], self.player[1] + d[1])
        reward = -0.1
        if not self._blocked(np_):
            if np_ in self.boxes:
                nb = (np_[0] + d[0], np_[1] + d[1])
                if not self._blocked(nb) and nb not in self.boxes:
                    was = np_ in self.targets
                    now = nb in self.targets
                    self.boxes.remove(np_)
                    self.boxes.add(nb)
                    reward += (1.0 if (now and not was) else 0.0) - (1.0 if (was and not now) else 0.0)
                    self.player = np_
            else:
                self.player = np_
        self._step_count += 1
        solved = self.boxes == self.targets
        done = solved or self._step_count >= self._max_step
        if solved:
            reward += 10.0
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
        return f"SokobanEnv({self.n}, boxes={self.num_boxes})"
