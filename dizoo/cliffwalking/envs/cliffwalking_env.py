"""CliffWalking-v0 implemented natively (exact gym dynamics): 4x12 grid,
start (3,0), goal (3,11), cells (3,1..10) are the cliff (-100, teleport to
start); reward -1 per step; obs one-hot 48; actions up/right/down/left.
Parity: reference dizoo/cliffwalking/envs/cliffwalking_env.py
(cliffwalking_dqn_config.py stop_value -13 = optimal)."""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('cliffwalking')
class CliffWalkingEnv(BaseEnv):

    ROWS, COLS = 4, 12

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_step = self._cfg.get('max_step', 300)
        self._observation_space = Box(0.0, 1.0, (48, ))
        self._action_space = Discrete(4)  # 0 up, 1 right, 2 down, 3 left
        self._reward_space = Box(-100.0, 0.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        self._pos = (3, 0)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        v = np.zeros(48, dtype=np.float32)
        v[self._pos[0] * self.COLS + self._pos[1]] = 1.0
        return v

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.item())
        r, c = self._pos
        dr, dc = [(-1, 0), (0, 1), (1, 0), (0, -1)][int(action)]
        r = int(np.clip(r + dr, 0, self.ROWS - 1))
        c = int(np.clip(c + dc, 0, self.COLS - 1))
        reward, done = -1.0, False
        if r == 3 and 1 <= c <= 10:  # cliff
            reward = -100.0
            r, c = 3, 0
        elif (r, c) == (3, 11):
            done = True
        self._pos = (r, c)
        self._step_count += 1
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
        return "CliffWalkingEnv"
