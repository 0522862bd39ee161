"""Cross-node context/model exchange over the event bus.

Parity: reference ding/framework/middleware/distributer.py
(ContextExchanger:13, ModelExchanger:198, PeriodicalModelExchanger:293).
"""
import logging
import time
from typing import Any, Dict, List, Optional

import torch

from ding.data import StorageLoader, Storage, ModelLoader
from ..context import OnlineRLContext
from ..task import task, Role

logger = logging.getLogger('ding')


class ContextExchanger:
    """Role-aware ctx merge: collectors emit trajectories/env_step after each
    iteration; the learner emits train_iter. Payload handlers follow the
    ``_put_<key>`` / ``_fetch_<key>`` naming convention."""

    def __init__(self, skip_n_iter: int = 1, storage_loader: Optional[StorageLoader] = None) -> None:
        if not task.router.is_active:
            raise RuntimeError("ContextExchanger requires an active task router (ditask)")
        if len(task.roles) == 0:
            logger.warning("no role on this node, exchanger is void")
        self._state: Dict[str, Any] = {}
        self._local_state: Dict[str, Any] = {}
        self._event_name = "context_exchanger_{role}"
        self._skip_n_iter = skip_n_iter
        self._storage_loader = storage_loader
        for role in task.role:
            if task.has_role(role):
                task.on(self._event_name.format(role=role), self.put)

    def __new__(cls, *args, **kwargs):
        if not task.router.is_active:
            return task.void()
        return super().__new__(cls)

    def __call__(self, ctx: OnlineRLContext):
        self.merge(ctx)
        yield
        payload = self.fetch(ctx)
        if payload:
            if self._storage_loader and task.has_role(task.role.COLLECTOR):
                payload = self._storage_loader.save(payload)
            # address the PEER roles' topics (a node listens on its own
            # role topic; emitting to our own topic would loop back to us)
            for role in task.role:
                if not task.has_role(role):
                    task.emit(self._event_name.format(role=role), payload, only_remote=True)

    def __del__(self):
        if self._storage_loader is not None:
            self._storage_loader.shutdown()

    def put(self, payload: Any):
        def _put(data: Dict):
            for key, item in data.items():
                fn_name = "_put_" + key
                if hasattr(self, fn_name):
                    getattr(self, fn_name)(item)

        if isinstance(payload, Storage):
            assert self._storage_loader is not None, "storage payload without a storage loader"
            self._storage_loader.load(payload, _put)
        else:
            _put(payload)

    def fetch(self, ctx: OnlineRLContext) -> Dict[str, Any]:
        payload = {}
        for key, item in ctx.items():
            fn_name = "_fetch_" + key
            if hasattr(self, fn_name):
                value = getattr(self, fn_name)(item)
                if value is not None:
                    payload[key] = value
        return payload

    def merge(self, ctx: OnlineRLContext):
        if task.has_role(task.role.LEARNER):
            # learner blocks until trajectories arrive
            while len(self._state) < 1 and not task.finish:
                time.sleep(0.01)
        else:
            # collectors wait briefly for model/train_iter updates
            if ctx.total_step >= self._skip_n_iter:
                start = time.time()
                while len(self._state) < 1 and not task.finish:
                    if time.time() - start > 60:
                        break
                    time.sleep(0.01)
        for k, v in self._state.items():
            if k.startswith('_acc_'):
                continue
            ctx[k] = v
        # accumulate trajectories from several collectors
        if '_acc_trajectories' in self._state and self._state['_acc_trajectories']:
            ctx.trajectories = self._state['_acc_trajectories']
        if '_acc_episodes' in self._state and self._state['_acc_episodes']:
            ctx.episodes = self._state['_acc_episodes']
        self._state = {}

    # ------------------------------------------------- put/fetch handlers
    # collector -> learner
    def _put_trajectories(self, traj: List[Any]):
        if not task.has_role(task.role.LEARNER):
            return
        self._state.setdefault('_acc_trajectories', []).extend(traj)

    def _fetch_trajectories(self, traj: List[Any]):
        if task.has_role(task.role.COLLECTOR):
            return traj

    def _put_episodes(self, episodes: List[Any]):
        if not task.has_role(task.role.LEARNER):
            return
        self._state.setdefault('_acc_episodes', []).extend(episodes)

    def _fetch_episodes(self, episodes: List[Any]):
        if task.has_role(task.role.COLLECTOR):
            return episodes

    def _put_trajectory_end_idx(self, idx: List[int]):
        if not task.has_role(task.role.LEARNER):
            return
        self._state.setdefault('trajectory_end_idx', []).extend(idx)

    def _fetch_trajectory_end_idx(self, idx: List[int]):
        if task.has_role(task.role.COLLECTOR):
            return idx

    def _put_env_step(self, env_step: int):
        if not task.has_role(task.role.COLLECTOR):
            self._state['env_step'] = env_step

    def _fetch_env_step(self, env_step: int):
        if task.has_role(task.role.COLLECTOR):
            return env_step

    def _put_env_episode(self, env_episode: int):
        if not task.has_role(task.role.COLLECTOR):
            self._state['env_episode'] = env_episode

    def _fetch_env_episode(self, env_episode: int):
        if task.has_role(task.role.COLLECTOR):
            return env_episode

    # learner -> collectors/evaluators
    def _put_train_iter(self, train_iter: int):
        if not task.has_role(task.role.LEARNER):
            self._state['train_iter'] = train_iter

    def _fetch_train_iter(self, train_iter: int):
        if task.has_role(task.role.LEARNER):
            return train_iter


class ModelExchanger:
    """Learner broadcasts state_dict after each iteration; collectors load
    the freshest copy before inference."""

    def __init__(self, model: torch.nn.Module, model_loader: Optional[ModelLoader] = None) -> None:
        self._model = model
        self._model_loader = model_loader
        self._event_name = "model_exchanger"
        self._state_dict_cache: Optional[Any] = None
        self._is_learner = task.has_role(task.role.LEARNER)
        if not self._is_learner:
            task.on(self._event_name, self._cache_state_dict)
        if model_loader:
            task.once("finish", lambda *a, **k: model_loader.shutdown())

    def __new__(cls, *args, **kwargs):
        if not task.router.is_active:
            return task.void()
        if len(task.roles) == 0:
            return task.void()
        return super().__new__(cls)

    def _cache_state_dict(self, state_dict):
        self._state_dict_cache = state_dict

    def __call__(self, ctx) -> Any:
        if self._is_learner:
            yield
            self._send_model()
        else:
            self._update_model()

    def _update_model(self):
        # non-blocking: run with the current (possibly stale) weights until
        # the learner's broadcast lands — blocking here deadlocks against the
        # learner, which waits for trajectories before its first send
        if self._state_dict_cache is None or task.finish:
            return
        sd = self._state_dict_cache
        self._state_dict_cache = None
        if isinstance(sd, Storage) and self._model_loader is not None:
            sd = self._model_loader.load(sd)
        self._model.load_state_dict(sd)

    def _send_model(self):
        if self._model_loader:
            self._model_loader.save(self._send_callback)
        else:
            sd = {k: v.detach().cpu() for k, v in self._model.state_dict().items()}
            task.emit(self._event_name, sd, only_remote=True)

    def _send_callback(self, storage: Storage):
        if task.running:
            task.emit(self._event_name, storage, only_remote=True)


class PeriodicalModelExchanger(ModelExchanger):
    """ModelExchanger with send/update period control."""

    def __init__(self, model, mode: str = "fetch", period: int = 1, delay_toleration: float = float('inf'),
                 model_loader=None) -> None:
        super().__init__(model, model_loader)
        self._period = period
        self._mode = mode
        self._count = 0

    def __call__(self, ctx) -> Any:
        if self._is_learner:
            yield
            if self._count % self._period == 0:
                self._send_model()
            self._count += 1
        else:
            if self._count % self._period == 0:
                self._update_model()
            self._count += 1
