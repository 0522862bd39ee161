"""Serial (in-process) vectorized env manager with per-env state machine,
retry and watchdog timeouts.

Parity: reference ding/envs/env_manager/base_env_manager.py (EnvState:20,
BaseEnvManager:64, BaseEnvManagerV2:570).
"""
import copy
import enum
import logging
import platform
import time
import traceback
from collections import namedtuple
from functools import partial
from types import MethodType
from typing import Any, Callable, Dict, List, Optional, Union

import numpy as np
import torch

from ding.utils import ENV_MANAGER_REGISTRY, EasyDict, WatchDog, deep_merge_dicts
from ..env.base_env import BaseEnvTimestep

logger = logging.getLogger('ding')


class EnvState(enum.IntEnum):
    VOID = 0
    INIT = 1
    RUN = 2
    RESET = 3
    DONE = 4
    ERROR = 5


def timeout_wrapper(func: Callable, timeout: Optional[int]) -> Callable:
    if timeout is None or platform.system() == 'Windows':
        return func

    def wrapper(*args, **kwargs):
        watchdog = WatchDog(timeout)
        try:
            watchdog.start()
            return func(*args, **kwargs)
        finally:
            watchdog.stop()

    return wrapper


def retry_wrapper(func: Callable, max_retry: int, retry_waiting_time: float) -> Callable:

    def wrapper(*args, **kwargs):
        exceptions = []
        for _ in range(max_retry):
            try:
                return func(*args, **kwargs)
            except BaseException as e:
                exceptions.append(e)
                time.sleep(retry_waiting_time)
        raise RuntimeError(f"env method failed after {max_retry} retries: {exceptions[-1]}") from exceptions[-1]

    return wrapper


@ENV_MANAGER_REGISTRY.register('base')
class BaseEnvManager:
    """Serial vectorization: envs run in the caller's process."""

    config = dict(
        episode_num=float("inf"),
        max_retry=1,
        retry_type='reset',
        auto_reset=True,
        step_timeout=None,
        reset_timeout=None,
        retry_waiting_time=0.1,
    )

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))

    def __init__(self, env_fn: List[Callable], cfg: EasyDict = EasyDict({})):
        self._cfg = deep_merge_dicts(EasyDict(self.config), cfg)
        self._env_fn = env_fn
        self._env_num = len(env_fn)
        self._closed = True
        self._env_replay_path = None
        self._env_ref = self._env_fn[0]()
        try:
            self._observation_space = self._env_ref.observation_space
            self._action_space = self._env_ref.action_space
            self._reward_space = getattr(self._env_ref, 'reward_space', None)
        except Exception:
            self._observation_space = self._action_space = self._reward_space = None
        self._env_states: Dict[int, EnvState] = {i: EnvState.VOID for i in range(self._env_num)}
        self._env_seed: Dict[int, Optional[int]] = {i: None for i in range(self._env_num)}
        self._episode_num = self._cfg.episode_num
        self._max_retry = max(self._cfg.max_retry, 1)
        self._auto_reset = self._cfg.auto_reset
        self._retry_type = self._cfg.retry_type
        self._step_timeout = self._cfg.step_timeout
        self._reset_timeout = self._cfg.reset_timeout
        self._retry_waiting_time = self._cfg.retry_waiting_time
        self._env_episode_count = {i: 0 for i in range(self._env_num)}
        self._ready_obs = {}
        self._reset_param = {}

    @property
    def env_num(self) -> int:
        return self._env_num

    @property
    def env_ref(self):
        return self._env_ref

    @property
    def observation_space(self):
        return self._observation_space

    @property
    def action_space(self):
        return self._action_space

    @property
    def reward_space(self):
        return self._reward_space

    @property
    def ready_obs(self) -> Dict[int, Any]:
        """{env_id: obs} for envs currently in RUN state."""
        active = [i for i, s in self._env_states.items() if s == EnvState.RUN]
        return {i: self._ready_obs[i] for i in active}

    @property
    def ready_obs_id(self) -> List[int]:
        return [i for i, s in self._env_states.items() if s == EnvState.RUN]

    @property
    def ready_imgs(self):
        raise NotImplementedError

    @property
    def done(self) -> bool:
        return all(c >= self._episode_num for c in self._env_episode_count.values())

    @property
    def closed(self) -> bool:
        return self._closed

    @property
    def method_name_list(self) -> list:
        return ['reset', 'step', 'seed', 'close', 'enable_save_replay']

    def env_state_done(self, env_id: int) -> bool:
        return self._env_states[env_id] == EnvState.DONE

    def launch(self, reset_param: Optional[Dict] = None) -> None:
        assert self._closed, "launch() requires a closed manager"
        self._create_state()
        self.reset(reset_param)

    def _create_state(self) -> None:
        self._env_episode_count = {i: 0 for i in range(self.env_num)}
        self._ready_obs = {i: None for i in range(self.env_num)}
        self._envs = [fn() for fn in self._env_fn]
        assert len(self._envs) == self._env_num
        self._reset_param = {i: {} for i in range(self.env_num)}
        self._env_states = {i: EnvState.INIT for i in range(self.env_num)}
        if self._env_replay_path is not None:
            for e, path in zip(self._envs, self._env_replay_path):
                e.enable_save_replay(path)
        self._closed = False

    def reset(self, reset_param: Optional[Dict] = None) -> None:
        if reset_param is None:
            reset_param = {i: {} for i in range(self.env_num)}
        self._reset_param.update(reset_param)
        for env_id in reset_param:
            self._reset(env_id)

    def _reset(self, env_id: int) -> None:

        def reset_fn():
            if self._env_seed[env_id] is not None:
                if self._env_dynamic_seed is not None:
                    self._envs[env_id].seed(self._env_seed[env_id], self._env_dynamic_seed)
                else:
                    self._envs[env_id].seed(self._env_seed[env_id])
                self._env_seed[env_id] = None
            obs = self._envs[env_id].reset(**self._reset_param[env_id])
            self._ready_obs[env_id] = obs
            self._env_states[env_id] = EnvState.RUN

        fn = timeout_wrapper(reset_fn, self._reset_timeout)
        exceptions = []
        for _ in range(self._max_retry):
            try:
                self._env_states[env_id] = EnvState.RESET
                fn()
                return
            except BaseException as e:
                if self._retry_type == 'renew':
                    self._envs[env_id].close()
                    self._envs[env_id] = self._env_fn[env_id]()
                exceptions.append(e)
                time.sleep(self._retry_waiting_time)
        self._env_states[env_id] = EnvState.ERROR
        self.close()
        logger.error(f"env {env_id} reset failed {self._max_retry} times")
        raise RuntimeError(f"env {env_id} reset error: {exceptions[-1]}") from exceptions[-1]

    def step(self, actions: Dict[int, Any]) -> Dict[int, BaseEnvTimestep]:
        timesteps = {}
        for env_id, act in actions.items():
            timesteps[env_id] = self._step(env_id, act)
            if timesteps[env_id].done:
                self._env_episode_count[env_id] += 1
                if self._env_episode_count[env_id] < self._episode_num and self._auto_reset:
                    self._reset(env_id)
                else:
                    self._env_states[env_id] = EnvState.DONE
            else:
                self._ready_obs[env_id] = timesteps[env_id].obs
        return timesteps

    def _step(self, env_id: int, act: Any) -> BaseEnvTimestep:

        def step_fn():
            return self._envs[env_id].step(act)

        fn = timeout_wrapper(step_fn, self._step_timeout)
        exceptions = []
        for _ in range(self._max_retry):
            try:
                return fn()
            except BaseException as e:
                exceptions.append(e)
        self._env_states[env_id] = EnvState.ERROR
        logger.error(f"env {env_id} step failed {self._max_retry} times")
        raise RuntimeError(f"env {env_id} step error: {exceptions[-1]}") from exceptions[-1]

    def seed(self, seed: Union[Dict[int, int], List[int], int], dynamic_seed: Optional[bool] = None) -> None:
        if isinstance(seed, (int, np.integer)):
            seed = [seed + i for i in range(self.env_num)]
        if isinstance(seed, list):
            assert len(seed) == self._env_num
            seed = {i: s for i, s in enumerate(seed)}
        self._env_seed = seed
        self._env_dynamic_seed = dynamic_seed

    def enable_save_replay(self, replay_path: Union[List[str], str]) -> None:
        if isinstance(replay_path, str):
            replay_path = [replay_path] * self.env_num
        self._env_replay_path = replay_path

    def close(self) -> None:
        if self._closed:
            return
        for env in self._envs:
            try:
                env.close()
            except Exception:
                pass
        for i in range(self._env_num):
            self._env_states[i] = EnvState.VOID
        self._closed = True

    def random_action(self) -> Dict[int, Any]:
        return {i: self._envs[i].random_action() for i in self.ready_obs_id}


@ENV_MANAGER_REGISTRY.register('base_v2')
class BaseEnvManagerV2(BaseEnvManager):
    """V2: ready_obs is a stacked tensor aligned with ready_obs_id; step takes
    a list of actions and returns a list of timesteps whose info carries
    env_id (new Task/Middleware pipeline interface)."""

    @property
    def ready_obs(self) -> torch.Tensor:
        active = self.ready_obs_id
        obs = [self._ready_obs[i] for i in active]
        from ding.torch_utils import to_tensor
        obs = to_tensor(obs)
        if isinstance(obs[0], dict):
            return {k: torch.stack([o[k] for o in obs]) for k in obs[0]}
        return torch.stack([o if isinstance(o, torch.Tensor) else torch.as_tensor(o) for o in obs])

    def step(self, actions: Union[List[Any], Dict[int, Any]]) -> List[BaseEnvTimestep]:
        if not isinstance(actions, dict):
            actions = {i: a for i, a in zip(self.ready_obs_id, actions)}
        out = super().step(actions)
        timesteps = []
        for env_id, ts in out.items():
            info = dict(ts.info or {})
            info['env_id'] = env_id
            timesteps.append(BaseEnvTimestep(ts.obs, ts.reward, ts.done, info))
        return timesteps


def create_env_manager(manager_cfg: EasyDict, env_fn: List[Callable]) -> BaseEnvManager:
    manager_cfg = copy.deepcopy(manager_cfg)
    if 'import_names' in manager_cfg:
        from ding.utils import import_module
        import_module(manager_cfg.pop('import_names'))
    manager_type = manager_cfg.pop('type')
    return ENV_MANAGER_REGISTRY.build(manager_type, env_fn=env_fn, cfg=manager_cfg)


def get_env_manager_cls(cfg: EasyDict) -> type:
    import ding.envs.env_manager.subprocess_env_manager  # ensure registration
    return ENV_MANAGER_REGISTRY.get(cfg.type)
