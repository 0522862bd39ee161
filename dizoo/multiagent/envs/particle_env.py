"""Cooperative particle env (MPE ``simple_spread``): N agents move on a 2D
plane to cover N landmarks while avoiding collisions. Self-contained NumPy
physics (double-integrator with damping), no external dependency — the
MI355X-native stand-in for the reference's multiagent_particle / PettingZoo
family (reference dizoo/petting_zoo/envs/petting_zoo_simple_spread_env.py).

Observation follows the MPE convention per agent: [self_vel(2), self_pos(2),
landmark_rel(2N), other_agents_rel(2(N-1))]; the SMAC-style dict layout
(agent_state / global_state / action_mask) keeps it drop-in for
QMIX/MAPPO/COMA pipelines. Discrete(5) actions: no-op / +-x / +-y thrust.
Team reward: -sum over landmarks of the distance to the nearest agent,
minus a collision penalty.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('particle_spread')
class ParticleSpreadEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        self.agent_num = cfg.get('agent_num', 3)
        self.landmark_num = cfg.get('landmark_num', self.agent_num)
        self.max_step = cfg.get('max_step', 25)
        self.collide_penalty = cfg.get('collide_penalty', 1.0)
        self.continuous_actions = cfg.get('continuous_actions', False)
        self.dt = 0.1
        self.damping = 0.25
        self.accel = 5.0
        self.agent_size = 0.15
        self.obs_dim = 4 + 2 * self.landmark_num + 2 * (self.agent_num - 1)
        self.global_dim = 4 * self.agent_num + 2 * self.landmark_num
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        self._observation_space = Box(-np.inf, np.inf, (self.agent_num, self.obs_dim))
        self._action_space = Discrete(5)
        self._reward_space = Box(-np.inf, 0, (1, ))

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> dict:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._pos = self._rng.uniform(-1, 1, size=(self.agent_num, 2)).astype(np.float32)
        self._vel = np.zeros((self.agent_num, 2), dtype=np.float32)
        self._landmarks = self._rng.uniform(-1, 1, size=(self.landmark_num, 2)).astype(np.float32)
        self._step = 0
        self._eval_episode_return = 0.0
        return self._get_obs()

    def _get_obs(self) -> dict:
        obs = np.zeros((self.agent_num, self.obs_dim), dtype=np.float32)
        for i in range(self.agent_num):
            parts = [self._vel[i], self._pos[i], (self._landmarks - self._pos[i]).reshape(-1)]
            others = np.delete(self._pos, i, axis=0) - self._pos[i]
            parts.append(others.reshape(-1))
            obs[i] = np.concatenate(parts)
        gs = np.concatenate([self._pos.reshape(-1), self._vel.reshape(-1),
                             self._landmarks.reshape(-1)]).astype(np.float32)
        return {
            'agent_state': obs,
            'global_state': gs,
            'action_mask': np.ones((self.agent_num, 5), dtype=np.float32),
        }

    _THRUST = np.array([[0, 0], [-1, 0], [1, 0], [0, -1], [0, 1]], dtype=np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        if self.continuous_actions:
            force = np.clip(np.asarray(action, dtype=np.float32).reshape(self.agent_num, 2), -1, 1) * self.accel
        else:
            action = np.asarray(action).reshape(-1).astype(np.int64)
            force = self._THRUST[action] * self.accel
        self._vel = self._vel * (1 - self.damping) + force * self.dt
        self._pos = self._pos + self._vel * self.dt
        # coverage reward: each landmark scored by its nearest agent
        d = np.linalg.norm(self._pos[None, :, :] - self._landmarks[:, None, :], axis=-1)
        reward = -float(d.min(axis=1).sum())
        # collision penalty between agent pairs
        for i in range(self.agent_num):
            for j in range(i + 1, self.agent_num):
                if np.linalg.norm(self._pos[i] - self._pos[j]) < 2 * self.agent_size:
                    reward -= self.collide_penalty
        self._step += 1
        self._eval_episode_return += reward
        done = self._step >= self.max_step
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._get_obs(), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        if self.continuous_actions:
            return self._rng.uniform(-1, 1, size=(self.agent_num, 2)).astype(np.float32)
        return self._rng.randint(0, 5, size=(self.agent_num, ))

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
        return f"ParticleSpreadEnv(n={self.agent_num})"
