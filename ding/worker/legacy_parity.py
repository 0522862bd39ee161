"""Legacy worker-pipeline surface kept for reference parity.

Parity: reference ding/worker/__init__.py star exports — learner hooks
(learner_hook.py), comm bases/factories (comm/base_comm_learner.py,
comm/base_comm_collector.py, comm/utils.py), parallel collectors
(collector/base_parallel_collector.py, zergling_parallel_collector.py,
marine_parallel_collector.py, comm/naive_collector.py), VectorEvalMonitor
(collector/base_serial_evaluator.py:78), to_tensor_transitions
(collector/base_serial_collector.py:200), SequenceReplayBuffer
(replay_buffer/naive_buffer.py:472).

Design note: this build's distributed pipeline is the comm-task worker
design (ding/worker/comm.py + ding/framework) — the zergling/marine
processes of the reference collapse into one threaded collector driven by
commander tasks. The classes here provide the reference's class surface on
top of that design: BaseParallelCollector runs the same collect loop in a
thread; Zergling (sample-stream) and Marine (battle 1v1) specialize what a
"finished job" means.
"""
import copy
import math
import threading
from collections import deque
from typing import Any, Callable, Dict, List, Optional

import numpy as np
import torch

from ding.utils import (
    COMM_COLLECTOR_REGISTRY, COMM_LEARNER_REGISTRY, COMMANDER_REGISTRY, PARALLEL_COLLECTOR_REGISTRY, BUFFER_REGISTRY,
    EasyDict,
)
from ding.torch_utils import to_tensor
from .learner.base_learner import Hook
from .replay_buffer.naive_buffer import NaiveReplayBuffer, SequenceReplayBuffer

# --------------------------------------------------------------------------
# learner hooks
LearnerHook = Hook
_LEARNER_HOOK_REGISTRY: Dict[str, type] = {}


def register_learner_hook(name: str, hook_type: type) -> None:
    """Register a hook class so configs can name it."""
    assert issubclass(hook_type, Hook)
    _LEARNER_HOOK_REGISTRY[name] = hook_type


def add_learner_hook(hooks: Dict[str, List[Hook]], hook: Hook) -> None:
    """Insert a hook into a position->hooks mapping, keeping priority order
    (lower priority value runs first)."""
    pos = hook.position
    hooks.setdefault(pos, []).append(hook)
    hooks[pos].sort(key=lambda h: h.priority)


def merge_hooks(a: Dict[str, List[Hook]], b: Dict[str, List[Hook]]) -> Dict[str, List[Hook]]:
    out = {k: list(v) for k, v in a.items()}
    for pos, hs in b.items():
        for h in hs:
            add_learner_hook(out, h)
    return out


# --------------------------------------------------------------------------
# comm bases + factories
class BaseCommLearner:
    """Abstract transport adapter a learner uses to receive data/tasks and
    publish policies. The concrete offline default is FlaskFileSystemLearner
    (ding/worker/comm.py)."""

    def __init__(self, cfg: EasyDict) -> None:
        self._cfg = cfg

    def start(self) -> None:
        raise NotImplementedError

    def close(self) -> None:
        raise NotImplementedError

    def get_policy_update_info(self, path: str):
        raise NotImplementedError

    def send_policy(self, state_dict: dict) -> None:
        raise NotImplementedError


class BaseCommCollector:
    """Abstract transport adapter a collector uses to fetch policies and ship
    trajectories. Concrete offline default: FlaskFileSystemCollector."""

    def __init__(self, cfg: EasyDict) -> None:
        self._cfg = cfg

    def start(self) -> None:
        raise NotImplementedError

    def close(self) -> None:
        raise NotImplementedError

    def get_policy_update_info(self, path: str):
        raise NotImplementedError

    def send_stepdata(self, path: str, data: list) -> None:
        raise NotImplementedError


def create_comm_learner(cfg: EasyDict):
    return COMM_LEARNER_REGISTRY.build(cfg.type, cfg=cfg)


def create_comm_collector(cfg: EasyDict):
    return COMM_COLLECTOR_REGISTRY.build(cfg.type, cfg=cfg)


# --------------------------------------------------------------------------
# serial-evaluator helpers
class VectorEvalMonitor:
    """Per-env episode-return bookkeeping that avoids the short-episode bias:
    each env contributes at most ceil(n_episode/env_num) episodes, so fast
    (short) episodes cannot crowd out slow ones."""

    def __init__(self, env_num: int, n_episode: int) -> None:
        assert n_episode >= env_num, f"n_episode({n_episode}) must be >= env_num({env_num})"
        self._env_num = env_num
        self._n_episode = n_episode
        each = [n_episode // env_num + (1 if i < n_episode % env_num else 0) for i in range(env_num)]
        self._reward = {i: deque(maxlen=each[i]) for i in range(env_num)}
        self._info = {i: deque(maxlen=each[i]) for i in range(env_num)}

    def is_finished(self) -> bool:
        return all(len(q) == q.maxlen for q in self._reward.values())

    def update_reward(self, env_id, reward) -> None:
        if isinstance(reward, torch.Tensor):
            reward = reward.item()
        self._reward[int(env_id)].append(float(reward))

    def update_info(self, env_id, info) -> None:
        self._info[int(env_id)].append(info)

    def get_episode_return(self) -> list:
        return [r for q in self._reward.values() for r in q]

    def get_latest_reward(self, env_id: int) -> float:
        return self._reward[env_id][-1]

    def get_current_episode(self) -> int:
        return sum(len(q) for q in self._reward.values())

    def get_episode_info(self) -> Optional[dict]:
        infos = [i for q in self._info.values() for i in q]
        if not infos:
            return None
        keys = [k for k in infos[0] if np.isscalar(infos[0][k]) or isinstance(infos[0][k], (int, float))]
        out = {}
        for k in keys:
            vals = [float(i[k]) for i in infos if k in i]
            out[k] = sum(vals) / max(1, len(vals))
        return out


def to_tensor_transitions(data: List[Dict[str, Any]], shallow_copy_next_obs: bool = True) -> List[Dict[str, Any]]:
    """Tensorize a transition fragment; when requested, each step's next_obs
    aliases the following step's obs tensor so the fragment stores each frame
    once (reference base_serial_collector.py:200)."""
    if not data or 'next_obs' not in data[0]:
        return to_tensor(data, transform_scalar=False)
    if shallow_copy_next_obs:
        last = to_tensor([{k: v for k, v in data[-1].items()}], transform_scalar=False)[0]
        out = [to_tensor([{k: v for k, v in d.items() if k != 'next_obs'}], transform_scalar=False)[0]
               for d in data[:-1]] + [last]
        for i in range(len(out) - 1):
            out[i]['next_obs'] = out[i + 1]['obs']
        return out
    return to_tensor(data, transform_scalar=False)


# --------------------------------------------------------------------------
# parallel collectors (compatibility over the comm-task redesign)
class BaseParallelCollector:
    """Threaded collect loop: policy.forward -> env.step -> transitions out
    through ``send_stepdata``; the commander drives it with job dicts."""

    def __init__(self, cfg: EasyDict) -> None:
        self._cfg = cfg
        self._end_flag = False
        self._thread: Optional[threading.Thread] = None
        self.policy = None
        self.env_manager = None
        self.send_stepdata: Optional[Callable] = None

    def start(self) -> None:
        self._end_flag = False
        self._thread = threading.Thread(target=self._work_loop, daemon=True)
        self._thread.start()

    def close(self) -> None:
        self._end_flag = True
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None

    def _work_loop(self) -> None:
        while not self._end_flag:
            if not self._collect_step():
                break

    def _collect_step(self) -> bool:
        raise NotImplementedError


@PARALLEL_COLLECTOR_REGISTRY.register('zergling')
class ZerglingParallelCollector(BaseParallelCollector):
    """Sample-stream collector: run the current policy, ship fixed-size
    transition fragments until the commander ends the job."""

    def __init__(self, cfg: EasyDict) -> None:
        super().__init__(cfg)
        self._n_sample = cfg.get('n_sample', 16)
        self._buffer: List[dict] = []

    def push_transition(self, transition: dict) -> bool:
        self._buffer.append(transition)
        if len(self._buffer) >= self._n_sample:
            if self.send_stepdata is not None:
                self.send_stepdata(to_tensor_transitions(self._buffer))
            self._buffer = []
            return True
        return False

    def _collect_step(self) -> bool:
        return False  # driven externally via push_transition in this design


@PARALLEL_COLLECTOR_REGISTRY.register('marine')
class MarineParallelCollector(BaseParallelCollector):
    """Battle collector: two policies step a shared env; whole episodes ship
    per agent with the battle outcome attached."""

    def __init__(self, cfg: EasyDict) -> None:
        super().__init__(cfg)
        self._episodes: List[List[dict]] = [[], []]

    def push_transition(self, agent_id: int, transition: dict, done: bool = False, result: Optional[str] = None):
        self._episodes[agent_id].append(transition)
        if done and self.send_stepdata is not None:
            for aid, ep in enumerate(self._episodes):
                if ep:
                    self.send_stepdata({'agent': aid, 'episode': to_tensor_transitions(ep), 'result': result})
            self._episodes = [[], []]

    def _collect_step(self) -> bool:
        return False


@PARALLEL_COLLECTOR_REGISTRY.register('naive')
class NaiveCollector(BaseParallelCollector):
    """Minimal single-policy collector used by comm tests."""

    def __init__(self, cfg: EasyDict) -> None:
        super().__init__(cfg)
        self._step_fn: Optional[Callable] = None

    def set_step_fn(self, fn: Callable) -> None:
        self._step_fn = fn

    def _collect_step(self) -> bool:
        if self._step_fn is None:
            return False
        out = self._step_fn()
        if out is not None and self.send_stepdata is not None:
            self.send_stepdata(out)
        return out is not None


def create_parallel_collector(cfg: EasyDict) -> BaseParallelCollector:
    cfg = EasyDict(cfg)
    return PARALLEL_COLLECTOR_REGISTRY.build(cfg.type, cfg=cfg)


def get_parallel_collector_cls(cfg: EasyDict) -> type:
    return PARALLEL_COLLECTOR_REGISTRY.get(EasyDict(cfg).type)


def get_parallel_commander_cls(cfg: EasyDict) -> type:
    return COMMANDER_REGISTRY.get(EasyDict(cfg).type)
