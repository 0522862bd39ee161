"""bsuite diagnostic envs implemented natively (reference
dizoo/bsuite/envs/bsuite_env.py wraps deepmind/bsuite; offline we implement
the classic diagnostics directly, selected by env_id):

* deep_sea/N: NxN exploration chain — only the all-right path reaches the
  +1 treasure, each right move costs 0.01/N; obs is the one-hot grid.
* bandit/N: 1-step N-armed bandit with fixed arm means.
* memory_len/N: recall the first-step cue after N distractor steps
  (the standalone dizoo/memory env family mirrors this one).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('bsuite')
class BSuiteEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        env_id = self._cfg.get('env_id', 'deep_sea/10')
        self.game, _, size = env_id.partition('/')
        self.size = int(size or 10)
        if self.game == 'deep_sea':
            self._obs_dim = self.size * self.size
            self._act_n = 2
        elif self.game == 'bandit':
            self._obs_dim = 1
            self._act_n = self.size
        elif self.game == 'memory_len':
            self._obs_dim = 3
            self._act_n = 2
        else:
            raise KeyError(f'unknown bsuite game {self.game}')
        self._observation_space = Box(-1.0, 1.0, (self._obs_dim, ))
        self._action_space = Discrete(self._act_n)
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
        self._t = 0
        self._eval_episode_return = 0.0
        if self.game == 'deep_sea':
            self._row, self._col = 0, 0
            # action mapping flips per row-column like the real deep_sea
            self._flip = self._rng.binomial(1, 0.5, (self.size, self.size))
            return self._ds_obs()
        if self.game == 'bandit':
            self._means = np.linspace(0, 1, self.size)
            self._rng.shuffle(self._means)
            return np.zeros(1, dtype=np.float32)
        self._cue = int(self._rng.randint(0, 2))
        return np.array([1.0, 2 * self._cue - 1, 0.0], dtype=np.float32)

    def _ds_obs(self) -> np.ndarray:
        v = np.zeros(self.size * self.size, dtype=np.float32)
        if self._row < self.size:
            v[self._row * self.size + self._col] = 1.0
        return v

    def step(self, action: Any) -> BaseEnvTimestep:
        if hasattr(action, 'reshape'):
            action = int(np.asarray(action).reshape(-1)[0])
        action = int(action)
        done, reward = False, 0.0
        if self.game == 'deep_sea':
            go_right = action != self._flip[self._row, self._col]
            if go_right:
                reward -= 0.01 / self.size
                self._col = min(self._col + 1, self.size - 1)
            else:
                self._col = max(self._col - 1, 0)
            self._row += 1
            if self._row >= self.size:
                if self._col == self.size - 1:
                    reward += 1.0
                done = True
            obs = self._ds_obs()
        elif self.game == 'bandit':
            reward = float(self._rng.binomial(1, self._means[action]))
            done = True
            obs = np.zeros(1, dtype=np.float32)
        else:  # memory_len
            self._t += 1
            if self._t >= self.size:
                reward = 1.0 if action == self._cue else -1.0
                done = True
            obs = np.array([0.0, 0.0, self._t / self.size], dtype=np.float32)
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(obs, np.array([reward], dtype=np.float32), done, info)

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
        return f"BSuiteEnv({self.game}/{self.size})"
