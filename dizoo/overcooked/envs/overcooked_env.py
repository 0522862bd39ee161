"""Overcooked 2-agent coordination (reference dizoo/overcooked wrapping
overcooked_ai). Native simplified 'cramped room': two cooks on a grid must
fetch onions -> pot (3 onions cooks a soup) -> plate -> serve. Team reward
+20 per served soup, small shaping for useful handoffs. SMAC-style dict obs
(per-agent local features + global kitchen state + action mask) so the
cooperative MARL lane (QMIX/VDN/MAPPO) runs unchanged. Discrete(6):
up/down/left/right/stay/interact.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

# cramped-room layout: X wall, O onion pile, P pot, D dish pile, S serve
LAYOUT = [
    "XXPXX",
    "O...O",
    "X...X",
    "XD.SX",
]


@ENV_REGISTRY.register('overcooked')
class OvercookedLiteEnv(BaseEnv):

    H, W = 4, 5

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_step = self._cfg.get('max_step', 200)
        self.agent_num = 2
        self.obs_dim = 10
        self.global_dim = 2 * 4 + 2  # both agents (pos+held) + pot state
        self._observation_space = Box(0.0, 1.0, (self.agent_num, self.obs_dim))
        self._action_space = Discrete(6)
        self._reward_space = Box(0.0, 20.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> dict:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self.pos = [np.array([1, 1]), np.array([2, 3])]
        self.held = [0, 0]  # 0 none, 1 onion, 2 dish, 3 soup
        self.pot = 0        # onions in pot; 3 -> soup ready (then 4 = ready)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _tile(self, r, c) -> str:
        if 0 <= r < self.H and 0 <= c < self.W:
            return LAYOUT[r][c]
        return 'X'

    def _obs(self) -> dict:
        per = np.zeros((2, self.obs_dim), dtype=np.float32)
        for i in range(2):
            r, c = self.pos[i]
            per[i, 0] = r / self.H
            per[i, 1] = c / self.W
            per[i, 2 + self.held[i]] = 1.0         # held one-hot (4)
            per[i, 6] = self.pot / 4.0
            other = self.pos[1 - i]
            per[i, 7] = other[0] / self.H
            per[i, 8] = other[1] / self.W
            per[i, 9] = self.held[1 - i] / 3.0
        gs = np.concatenate([
            self.pos[0] / [self.H, self.W], [self.held[0] / 3.0],
            self.pos[1] / [self.H, self.W], [self.held[1] / 3.0],
            [self.pot / 4.0, self._step_count / self._max_step]
        ]).astype(np.float32)
        return {
            'agent_state': per,
            'global_state': gs,
            'action_mask': np.ones((2, 6), dtype=np.float32),
        }

    def _interact(self, i: int) -> float:
        r, c = self.pos[i]
        reward = 0.0
        for dr, dc in ((-1, 0), (1, 0), (0, -1), (0, 1)):
            t = self._tile(r + dr, c + dc)
            if t == 'O' and self.held[i] == 0:
                self.held[i] = 1
                return 0.1
            if t == 'P':
                if self.held[i] == 1 and self.pot < 3:
                    self.held[i] = 0
                    self.pot += 1
                    if self.pot == 3:
                        self.pot = 4  # instantly ready (no cook timer)
                    return 0.2
                if self.held[i] == 2 and self.pot == 4:
                    self.held[i] = 3
                    self.pot = 0
                    return 0.5
            if t == 'D' and self.held[i] == 0:
                self.held[i] = 2
                return 0.1
            if t == 'S' and self.held[i] == 3:
                self.held[i] = 0
                return 20.0
        return reward

    def step(self, action: Any) -> BaseEnvTimestep:
        acts = np.asarray(action).reshape(-1).astype(np.int64)
        reward = 0.0
        moves = [(-1, 0), (1, 0), (0, -1), (0, 1), (0, 0)]
        for i in range(2):
            a = int(acts[i])
            if a < 5:
                nr, nc = self.pos[i] + moves[a]
                blocked = self._tile(nr, nc) != '.' or (nr, nc) == tuple(self.pos[1 - i])
                if not blocked:
                    self.pos[i] = np.array([nr, nc])
            else:
                reward += self._interact(i)
        self._step_count += 1
        done = self._step_count >= self._max_step
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return self._rng.randint(0, 6, size=(2, ))

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
        return "OvercookedLiteEnv"
