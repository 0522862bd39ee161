"""Env manager built on the Supervisor child-process RPC, with crash
auto-restart.

Parity: reference ding/envs/env_manager/env_supervisor.py:37.
"""
import copy
from typing import Any, Callable, Dict, List, Optional

import numpy as np

from ding.framework.supervisor import Supervisor, ChildType, SendPayload
from ding.utils import ENV_MANAGER_REGISTRY, EasyDict, deep_merge_dicts
from ..env.base_env import BaseEnvTimestep
from .base_env_manager import EnvState


@ENV_MANAGER_REGISTRY.register('env_supervisor')
class EnvSupervisor(Supervisor):
    """Each env lives in a supervised child (process or thread); crashed
    children are restarted and the env re-reset."""

    config = dict(
        episode_num=float("inf"),
        max_retry=1,
        auto_reset=True,
        shared_memory=False,
        retry_waiting_time=0.1,
    )

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))

    def __init__(self, type_: ChildType = ChildType.PROCESS, env_fn: List[Callable] = None, cfg: EasyDict = None,
                 **kwargs):
        super().__init__(type_=type_)
        self._cfg = deep_merge_dicts(self.default_config(), cfg or EasyDict({}))
        self._env_fn = env_fn or []
        self._env_num = len(self._env_fn)
        self._closed = True
        for fn in self._env_fn:
            self.register(fn)
        self._env_states: Dict[int, EnvState] = {i: EnvState.VOID for i in range(self._env_num)}
        self._ready_obs: Dict[int, Any] = {}
        self._env_seed: Dict[int, Optional[int]] = {}
        self._env_ref = self._env_fn[0]() if self._env_fn else None

    @property
    def env_num(self) -> int:
        return self._env_num

    @property
    def closed(self) -> bool:
        return self._closed

    @property
    def ready_obs(self) -> Dict[int, Any]:
        return {i: o for i, o in self._ready_obs.items() if self._env_states[i] == EnvState.RUN}

    @property
    def ready_obs_id(self) -> List[int]:
        return [i for i, s in self._env_states.items() if s == EnvState.RUN]

    @property
    def done(self) -> bool:
        return all(s == EnvState.DONE for s in self._env_states.values())

    def launch(self, reset_param: Optional[Dict] = None) -> None:
        self.start_link()
        self._closed = False
        for i, seed in self._env_seed.items():
            if seed is not None:
                self.send(SendPayload(proc_id=i, method='seed', args=[seed]))
        payloads = [SendPayload(proc_id=i, method='reset') for i in range(self._env_num)]
        for p in payloads:
            self.send(p)
        results = self.recv_all(payloads)
        for i, r in enumerate(results):
            self._ready_obs[i] = r.data
            self._env_states[i] = EnvState.RUN

    def seed(self, seed, dynamic_seed: Optional[bool] = None) -> None:
        if isinstance(seed, int):
            seed = {i: seed + i for i in range(self._env_num)}
        elif isinstance(seed, list):
            seed = {i: s for i, s in enumerate(seed)}
        self._env_seed = seed

    def step(self, actions: Dict[int, Any]) -> Dict[int, BaseEnvTimestep]:
        payloads = [SendPayload(proc_id=i, method='step', args=[a]) for i, a in actions.items()]
        for p in payloads:
            self.send(p)
        out = {}
        results = self.recv_all(payloads, ignore_err=True)
        for p, r in zip(payloads, results):
            env_id = p.proc_id
            if r.err is not None:
                # restart crashed env child
                self._children[env_id].restart(self._recv_q)
                rp = SendPayload(proc_id=env_id, method='reset')
                self.send(rp)
                obs = self.recv_all([rp])[0].data
                self._ready_obs[env_id] = obs
                out[env_id] = BaseEnvTimestep(obs, np.array([0.0], dtype=np.float32), False, {'abnormal': True})
                continue
            ts: BaseEnvTimestep = r.data
            out[env_id] = ts
            if ts.done:
                if self._cfg.auto_reset:
                    rp = SendPayload(proc_id=env_id, method='reset')
                    self.send(rp)
                    self._ready_obs[env_id] = self.recv_all([rp])[0].data
                    self._env_states[env_id] = EnvState.RUN
                else:
                    self._env_states[env_id] = EnvState.DONE
            else:
                self._ready_obs[env_id] = ts.obs
        return out

    def close(self, timeout: float = 1.0) -> None:
        if self._closed:
            return
        self._closed = True
        self.shutdown(timeout=timeout)
