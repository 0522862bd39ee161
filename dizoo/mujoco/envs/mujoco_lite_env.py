"""Synthetic MuJoCo-shaped continuous-control env ("mujoco-lite").

MuJoCo binaries are unavailable offline; this env preserves the Hopper
interface (obs 11, act 3, episode 1000) with smooth linear-quadratic hidden
dynamics so SAC/TD3/DDPG configs run end to end (reference
dizoo/mujoco/envs/mujoco_env.py).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box
from ding.utils import ENV_REGISTRY


_PRESETS = {
    'Hopper-v3': dict(obs=11, act=3),
    'HalfCheetah-v3': dict(obs=17, act=6),
    'Walker2d-v3': dict(obs=17, act=6),
    'Ant-v3': dict(obs=111, act=8),
    'Humanoid-v3': dict(obs=376, act=17),
}


@ENV_REGISTRY.register('mujoco_lite')
class MujocoLiteEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        env_id = cfg.get('env_id', 'Hopper-v3')
        preset = _PRESETS.get(env_id, dict(obs=cfg.get('obs_dim', 11), act=cfg.get('act_dim', 3)))
        self.obs_dim, self.act_dim = preset['obs'], preset['act']
        self.max_step = cfg.get('max_step', 1000)
        self._observation_space = Box(-10, 10, (self.obs_dim, ))
        self._action_space = Box(-1, 1, (self.act_dim, ))
        self._reward_space = Box(-10, 10, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True
        self._A = None

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        # fixed random linear dynamics per episode family (stable spectral norm)
        rng = np.random.RandomState(12345)
        self._A = rng.randn(self.obs_dim, self.obs_dim) * 0.1
        self._A /= max(1.0, np.abs(np.linalg.eigvals(self._A)).max() / 0.95)
        self._B = rng.randn(self.obs_dim, self.act_dim) * 0.5
        self._w = rng.randn(self.obs_dim) / np.sqrt(self.obs_dim)
        self._state = self._rng.randn(self.obs_dim) * 0.1
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._state.astype(np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        action = np.clip(np.asarray(action, dtype=np.float64).reshape(-1), -1, 1)
        self._state = self._A @ self._state + self._B @ action + self._rng.randn(self.obs_dim) * 0.01
        self._state = np.clip(self._state, -10, 10)
        # forward-progress style reward: projection along w minus action cost
        reward = float(self._w @ self._state) + 1.0 - 0.1 * float((action ** 2).sum())
        self._step_count += 1
        self._eval_episode_return += reward
        unhealthy = np.abs(self._state).max() > 9.5
        done = self._step_count >= self.max_step or unhealthy
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(self._state.astype(np.float32), np.array([reward], dtype=np.float32), done, info)

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return self._action_space.sample()

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
        return "MujocoLiteEnv"
