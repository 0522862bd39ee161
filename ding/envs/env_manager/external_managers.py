"""Optional external-backend env managers.

Parity: reference ding/envs/env_manager/envpool_env_manager.py:21
(PoolEnvManager, C++ batched envs) and gym_vector_env_manager.py:17.
Both backends are optional imports offline; the classes are registered and
raise with clear messages when the package is absent.
"""
from typing import Any, Dict, List, Optional

import numpy as np

from ding.utils import ENV_MANAGER_REGISTRY, EasyDict
from ..env.base_env import BaseEnvTimestep


@ENV_MANAGER_REGISTRY.register('env_pool')
class PoolEnvManager:
    """envpool adapter: batched C++ envs with an async step interface."""

    config = dict(env_id='Pong-v5', env_num=8, batch_size=8)

    @classmethod
    def default_config(cls) -> EasyDict:
        import copy
        return EasyDict(copy.deepcopy(cls.config))

    def __init__(self, cfg: EasyDict):
        self._cfg = cfg
        try:
            import envpool
        except ImportError:
            raise ImportError(
                "envpool is not installed in this image; install it to use PoolEnvManager "
                "or switch env_manager.type to 'subprocess'"
            )
        self._envpool = envpool
        self._closed = True

    def launch(self) -> None:
        self._pool = self._envpool.make(
            self._cfg.env_id, env_type='gym', num_envs=self._cfg.env_num, batch_size=self._cfg.batch_size
        )
        self._pool.async_reset()
        self._closed = False

    @property
    def env_num(self) -> int:
        return self._cfg.env_num

    @property
    def ready_obs(self) -> Dict[int, Any]:
        obs, rew, done, info = self._pool.recv()
        self._last_ids = info['env_id']
        return {int(i): o for i, o in zip(info['env_id'], obs)}

    def step(self, actions: Dict[int, Any]) -> Dict[int, BaseEnvTimestep]:
        env_ids = np.array(list(actions.keys()), dtype=np.int32)
        acts = np.array(list(actions.values()))
        self._pool.send(acts, env_ids)
        obs, rew, done, info = self._pool.recv()
        return {
            int(i): BaseEnvTimestep(o, np.array([r], dtype=np.float32), bool(d), {})
            for i, o, r, d in zip(info['env_id'], obs, rew, done)
        }

    def close(self) -> None:
        self._closed = True


@ENV_MANAGER_REGISTRY.register('gym_vector')
class GymVectorEnvManager:
    """gym.vector backend (gym not installed offline; kept for parity)."""

    def __init__(self, env_fn: List, cfg: EasyDict):
        try:
            import gym
        except ImportError:
            raise ImportError(
                "gym is not installed in this image; use the in-repo BaseEnvManager/SubprocessEnvManager"
            )
