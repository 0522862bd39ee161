"""Cooperative matrix game: a lightweight self-contained MARL env (SMAC-style
observation dict {agent_state, global_state, action_mask}) used as the smoke
env for QMIX/WQMIX/COMA/MAPPO in place of the external SMAC binary.

Each episode draws a hidden target action pattern; the team reward each step
is the fraction of agents matching it. Observations encode the target
noisily so coordinated policies can learn it.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('coop_matrix')
class CoopMatrixEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        self.agent_num = cfg.get('agent_num', 3)
        self.action_dim = cfg.get('action_dim', 4)
        self.obs_dim = cfg.get('obs_dim', 8)
        self.max_step = cfg.get('max_step', 25)
        self.global_dim = self.agent_num * self.action_dim
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        self._observation_space = Box(-10, 10, (self.agent_num, self.obs_dim))
        self._action_space = Discrete(self.action_dim)
        self._reward_space = Box(0, 1, (1, ))

    def reset(self) -> dict:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._target = self._rng.randint(0, self.action_dim, size=(self.agent_num, ))
        self._step = 0
        self._eval_episode_return = 0.0
        return self._get_obs()

    def _get_obs(self) -> dict:
        # each agent sees a noisy one-hot of its own target + its id
        obs = np.zeros((self.agent_num, self.obs_dim), dtype=np.float32)
        for i in range(self.agent_num):
            obs[i, self._target[i] % self.obs_dim] = 1.0
            obs[i, (self.action_dim + i) % self.obs_dim] += 0.5
        obs += self._rng.randn(self.agent_num, self.obs_dim).astype(np.float32) * 0.01
        gs = np.zeros(self.global_dim, dtype=np.float32)
        for i in range(self.agent_num):
            gs[i * self.action_dim + self._target[i]] = 1.0
        return {
            'agent_state': obs,
            'global_state': gs,
            'action_mask': np.ones((self.agent_num, self.action_dim), dtype=np.float32),
        }

    def step(self, action: Any) -> BaseEnvTimestep:
        action = np.asarray(action).reshape(-1)
        match = (action == self._target).sum() / self.agent_num
        reward = float(match)
        self._step += 1
        self._eval_episode_return += reward
        done = self._step >= self.max_step
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(self._get_obs(), np.array([reward], dtype=np.float32), done, info)

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return self._rng.randint(0, self.action_dim, size=(self.agent_num, ))

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
        return "CoopMatrixEnv"
