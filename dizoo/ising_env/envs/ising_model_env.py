"""Ising-model MARL env (reference dizoo/ising_env/envs/ising_model_env.py):
N spins on a ring; each agent observes its neighbourhood spins and flips or
keeps its own spin; shared reward = alignment (mean-field order parameter).
SMAC-style obs dict so QMIX-family policies run unchanged.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('ising_model')
class IsingModelEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.agent_num = int(self._cfg.get('num_agents', 10))
        self.k = int(self._cfg.get('agent_view_sight', 2))  # neighbours each side
        self._max_step = self._cfg.get('max_step', 50)
        self.obs_dim = 2 * self.k + 1
        self.global_dim = self.agent_num
        self._observation_space = Box(-1.0, 1.0, (self.agent_num, self.obs_dim))
        self._action_space = Discrete(2)  # keep / flip
        self._reward_space = Box(-1.0, 1.0, (1, ))
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
        self.spins = self._rng.choice([-1.0, 1.0], size=self.agent_num).astype(np.float32)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> dict:
        per = np.zeros((self.agent_num, self.obs_dim), dtype=np.float32)
        for i in range(self.agent_num):
            idx = [(i + d) % self.agent_num for d in range(-self.k, self.k + 1)]
            per[i] = self.spins[idx]
        return {
            'agent_state': per,
            'global_state': self.spins.copy(),
            'action_mask': np.ones((self.agent_num, 2), dtype=np.float32),
        }

    def step(self, action: Any) -> BaseEnvTimestep:
        a = np.asarray(action).reshape(-1).astype(np.int64)
        flip = a == 1
        self.spins[flip] *= -1
        order = abs(float(self.spins.mean()))  # magnetisation in [0, 1]
        reward = order - 0.01 * float(flip.sum()) / self.agent_num
        self._step_count += 1
        done = self._step_count >= self._max_step or order == 1.0
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return self._rng.randint(0, 2, size=(self.agent_num, ))

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
        return f"IsingModelEnv(N={self.agent_num})"
