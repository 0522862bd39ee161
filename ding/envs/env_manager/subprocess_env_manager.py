"""Subprocess vectorized env manager: one child process per env, Pipe control
plane, shared-memory observation buffers.

Parity: reference ding/envs/env_manager/subprocess_env_manager.py
(SyncSubprocessEnvManager:678, AsyncSubprocessEnvManager:35,
SubprocessEnvManagerV2:779; shm path data/shm_buffer.py:52,66).
"""
import logging
import multiprocessing as mp
import pickle
import platform
import time
import traceback
from typing import Any, Callable, Dict, List, Optional, Union

import numpy as np
import torch

from ding.utils import ENV_MANAGER_REGISTRY, EasyDict, deep_merge_dicts
from ding.data.shm_buffer import ShmBuffer, ShmBufferContainer
from ..env.base_env import BaseEnvTimestep
from .base_env_manager import BaseEnvManager, EnvState

logger = logging.getLogger('ding')


def _worker_loop(child_conn, env_fn, shm_buffer, auto_reset: bool):
    """Child process: serve reset/step/seed/close/method commands."""
    env = env_fn()

    def _try_shm(obs):
        """Best-effort shm transport: envs whose obs doesn't match the
        declared observation_space (e.g. MARL dict obs) fall back to the
        pipe instead of crashing the worker."""
        if shm_buffer is None:
            return False
        try:
            shm_buffer.fill(np.ascontiguousarray(obs))
            return True
        except Exception:
            return False
    try:
        while True:
            cmd, payload = child_conn.recv()
            try:
                if cmd == 'reset':
                    obs = env.reset(**payload)
                    if _try_shm(obs):
                        child_conn.send(('ok', ('shm', )))
                    else:
                        child_conn.send(('ok', obs))
                elif cmd == 'step':
                    ts = env.step(payload)
                    obs = ts.obs
                    done = ts.done
                    if done and auto_reset:
                        new_obs = env.reset()
                    else:
                        new_obs = None
                    out_obs = new_obs if (done and auto_reset) else obs
                    if _try_shm(out_obs):
                        child_conn.send(('ok', ('shm', ts.reward, done, ts.info)))
                    else:
                        child_conn.send(('ok', (out_obs, ts.reward, done, ts.info)))
                elif cmd == 'seed':
                    env.seed(*payload)
                    child_conn.send(('ok', None))
                elif cmd == 'method':
                    name, args, kwargs = payload
                    ret = getattr(env, name)(*args, **kwargs)
                    child_conn.send(('ok', ret))
                elif cmd == 'attr':
                    child_conn.send(('ok', getattr(env, payload)))
                elif cmd == 'close':
                    env.close()
                    child_conn.send(('ok', None))
                    break
                else:
                    child_conn.send(('err', f"unknown cmd {cmd}"))
            except BaseException as e:
                child_conn.send(('err', f"{type(e).__name__}: {e}\n{traceback.format_exc()}"))
    finally:
        try:
            child_conn.close()
        except Exception:
            pass


@ENV_MANAGER_REGISTRY.register('subprocess')
class SyncSubprocessEnvManager(BaseEnvManager):
    """Synchronous: every step() call drives the requested env ids and waits
    for all of them. Observations travel through shared memory when
    ``shared_memory=True`` and the env exposes a fixed obs shape."""

    config = dict(
        episode_num=float("inf"),
        max_retry=1,
        retry_type='renew',
        auto_reset=True,
        step_timeout=None,
        reset_timeout=None,
        retry_waiting_time=0.1,
        shared_memory=True,
        context='fork' if platform.system().lower() != 'windows' else 'spawn',
        wait_num=float("inf"),
        step_wait_timeout=None,
    )

    def __init__(self, env_fn: List[Callable], cfg: EasyDict = EasyDict({})):
        super().__init__(env_fn, cfg)
        self._shared_memory = self._cfg.shared_memory
        self._context = self._cfg.context
        self._pipes: List[Any] = []
        self._procs: List[Any] = []
        self._shm: List[Any] = []

    def _obs_spec(self):
        """Infer (dtype, shape) of the obs from the reference env."""
        space = self._observation_space
        if space is not None and getattr(space, 'shape', None):
            return np.dtype(space.dtype if space.dtype is not None else np.float32), tuple(space.shape)
        return None

    def _create_state(self) -> None:
        self._env_episode_count = {i: 0 for i in range(self.env_num)}
        self._ready_obs = {i: None for i in range(self.env_num)}
        self._reset_param = {i: {} for i in range(self.env_num)}
        ctx = mp.get_context(self._context)
        spec = self._obs_spec() if self._shared_memory else None
        self._pipes, self._procs, self._shm = [], [], []
        for i in range(self.env_num):
            shm = ShmBufferContainer(spec[0], spec[1], copy_on_get=True) if spec is not None else None
            parent, child = ctx.Pipe()
            proc = ctx.Process(
                target=_worker_loop, args=(child, self._env_fn[i], shm, self._auto_reset), daemon=True
            )
            proc.start()
            child.close()
            self._pipes.append(parent)
            self._procs.append(proc)
            self._shm.append(shm)
        self._env_states = {i: EnvState.INIT for i in range(self.env_num)}
        self._closed = False

    def _renew_env(self, env_id: int):
        try:
            self._procs[env_id].terminate()
        except Exception:
            pass
        ctx = mp.get_context(self._context)
        spec = self._obs_spec() if self._shared_memory else None
        shm = ShmBufferContainer(spec[0], spec[1], copy_on_get=True) if spec is not None else None
        parent, child = ctx.Pipe()
        proc = ctx.Process(target=_worker_loop, args=(child, self._env_fn[env_id], shm, self._auto_reset), daemon=True)
        proc.start()
        child.close()
        self._pipes[env_id] = parent
        self._procs[env_id] = proc
        self._shm[env_id] = shm

    def _recv(self, env_id: int, timeout: Optional[float] = None):
        if timeout is not None:
            if not self._pipes[env_id].poll(timeout):
                raise TimeoutError(f"env {env_id} ipc timeout ({timeout}s)")
        status, payload = self._pipes[env_id].recv()
        if status != 'ok':
            raise RuntimeError(f"env {env_id} subprocess error: {payload}")
        return payload

    def _get_obs(self, env_id: int, payload):
        if isinstance(payload, tuple) and len(payload) >= 1 and isinstance(payload[0], str) and payload[0] == 'shm':
            return self._shm[env_id].get()
        return payload

    def reset(self, reset_param: Optional[Dict] = None) -> None:
        if reset_param is None:
            reset_param = {i: {} for i in range(self.env_num)}
        self._reset_param.update(reset_param)
        ids = list(reset_param.keys())
        for env_id in ids:
            if self._env_seed.get(env_id) is not None:
                args = (self._env_seed[env_id], ) if self._env_dynamic_seed is None \
                    else (self._env_seed[env_id], self._env_dynamic_seed)
                self._pipes[env_id].send(('seed', args))
                self._recv(env_id)
                self._env_seed[env_id] = None
            self._env_states[env_id] = EnvState.RESET
            self._pipes[env_id].send(('reset', self._reset_param[env_id]))
        for env_id in ids:
            self._wait_reset(env_id)

    def _wait_reset(self, env_id: int):
        exceptions = []
        for attempt in range(self._max_retry):
            try:
                payload = self._recv(env_id, timeout=self._reset_timeout)
                obs = self._get_obs(env_id, payload)
                if not (isinstance(obs, tuple) and obs and isinstance(obs[0], str) and obs[0] == 'shm'):
                    self._ready_obs[env_id] = obs
                else:
                    self._ready_obs[env_id] = self._shm[env_id].get()
                self._env_states[env_id] = EnvState.RUN
                return
            except BaseException as e:
                exceptions.append(e)
                if attempt + 1 >= self._max_retry:
                    break
                if self._cfg.get('retry_type', 'renew') == 'renew' or isinstance(e, TimeoutError):
                    # hung/dead worker: replace the subprocess entirely
                    self._renew_env(env_id)
                else:
                    time.sleep(self._cfg.get('retry_waiting_time', 0.1))
                self._pipes[env_id].send(('reset', self._reset_param[env_id]))
        self._env_states[env_id] = EnvState.ERROR
        self.close()
        raise RuntimeError(f"env {env_id} reset failed: {exceptions[-1]}") from exceptions[-1]

    def step(self, actions: Dict[int, Any]) -> Dict[int, BaseEnvTimestep]:
        for env_id, act in actions.items():
            self._pipes[env_id].send(('step', act))
        timesteps = {}
        for env_id in actions:
            payload = self._recv(env_id, timeout=self._step_timeout)
            if isinstance(payload[0], str) and payload[0] == 'shm':
                _, reward, done, info = payload
                obs = self._shm[env_id].get()
            else:
                obs, reward, done, info = payload
            timesteps[env_id] = BaseEnvTimestep(obs, reward, done, info)
            if done:
                self._env_episode_count[env_id] += 1
                if self._env_episode_count[env_id] < self._episode_num and self._auto_reset:
                    # child already auto-reset; obs is the fresh reset obs
                    self._ready_obs[env_id] = obs
                    self._env_states[env_id] = EnvState.RUN
                else:
                    self._env_states[env_id] = EnvState.DONE
            else:
                self._ready_obs[env_id] = obs
        return timesteps

    def close(self) -> None:
        if self._closed:
            return
        for i, pipe in enumerate(self._pipes):
            try:
                pipe.send(('close', None))
            except Exception:
                pass
        time.sleep(0.1)
        for proc in self._procs:
            try:
                proc.terminate()
                proc.join(timeout=1)
            except Exception:
                pass
        for i in range(self._env_num):
            self._env_states[i] = EnvState.VOID
        self._closed = True

    def random_action(self) -> Dict[int, Any]:
        out = {}
        for env_id in self.ready_obs_id:
            self._pipes[env_id].send(('method', ('random_action', (), {})))
        for env_id in self.ready_obs_id:
            out[env_id] = self._recv(env_id)
        return out


@ENV_MANAGER_REGISTRY.register('async_subprocess')
class AsyncSubprocessEnvManager(SyncSubprocessEnvManager):
    """ready_obs returns whichever envs have finished their last step
    (wait_num / step_wait_timeout semantics)."""

    def __init__(self, env_fn: List[Callable], cfg: EasyDict = EasyDict({})):
        super().__init__(env_fn, cfg)
        self._pending: Dict[int, bool] = {}

    def _create_state(self) -> None:
        super()._create_state()
        self._pending = {}
        self._cached_timesteps = {}

    @property
    def ready_obs(self) -> Dict[int, Any]:
        self._poll_pending(block_for_one=True)
        active = [
            i for i, s in self._env_states.items() if s == EnvState.RUN and i not in self._pending
        ]
        return {i: self._ready_obs[i] for i in active}

    def _poll_pending(self, block_for_one: bool = False):
        while self._pending:
            progressed = False
            for env_id in list(self._pending.keys()):
                if self._pipes[env_id].poll(0):
                    payload = self._recv(env_id)
                    self._finish_step(env_id, payload)
                    del self._pending[env_id]
                    progressed = True
            if not block_for_one or progressed or not self._pending:
                break
            time.sleep(0.001)

    def _finish_step(self, env_id: int, payload):
        if isinstance(payload[0], str) and payload[0] == 'shm':
            _, reward, done, info = payload
            obs = self._shm[env_id].get()
        else:
            obs, reward, done, info = payload
        self._cached_timesteps[env_id] = BaseEnvTimestep(obs, reward, done, info)
        if done:
            self._env_episode_count[env_id] += 1
            if self._env_episode_count[env_id] < self._episode_num and self._auto_reset:
                self._ready_obs[env_id] = obs
                self._env_states[env_id] = EnvState.RUN
            else:
                self._env_states[env_id] = EnvState.DONE
        else:
            self._ready_obs[env_id] = obs

    def step(self, actions: Dict[int, Any]) -> Dict[int, BaseEnvTimestep]:
        for env_id, act in actions.items():
            self._pipes[env_id].send(('step', act))
            self._pending[env_id] = True
        # wait for at least wait_num or timeout
        wait_num = min(self._cfg.wait_num, len(self._pending))
        deadline = None if self._cfg.step_wait_timeout is None else time.time() + self._cfg.step_wait_timeout
        while len(self._cached_timesteps) < wait_num:
            self._poll_pending()
            if deadline is not None and time.time() > deadline:
                break
            time.sleep(0.0005)
        out = self._cached_timesteps
        self._cached_timesteps = {}
        return out


@ENV_MANAGER_REGISTRY.register('subprocess_v2')
class SubprocessEnvManagerV2(SyncSubprocessEnvManager):
    """V2 interface over subprocess workers (stacked ready_obs, list step)."""

    @property
    def ready_obs(self):
        active = self.ready_obs_id
        obs = [self._ready_obs[i] for i in active]
        if isinstance(obs[0], dict):
            return {k: torch.stack([torch.as_tensor(o[k]) for o in obs]) for k in obs[0]}
        return torch.stack([torch.as_tensor(np.ascontiguousarray(o)) for o in obs])

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
