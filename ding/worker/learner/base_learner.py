"""Train-loop engine with hook stages.

Parity: reference ding/worker/learner/base_learner.py (BaseLearner:17,
train:215) + learner_hook.py (hook registry and the standard
load/save-ckpt + log hooks).
"""
import copy
import os
from collections import deque
from typing import Any, Callable, Dict, List, Optional, Union

import numpy as np
import torch

from ding.utils import LEARNER_REGISTRY, EasyDict, build_logger, deep_merge_dicts, get_rank, save_file, read_file, \
    EasyTimer


class Hook:

    def __init__(self, name: str, priority: int = 100, position: str = 'after_iter'):
        self.name = name
        self.priority = priority
        self.position = position

    def __call__(self, engine: 'BaseLearner') -> None:
        raise NotImplementedError


class LambdaHook(Hook):

    def __init__(self, name, fn, priority=100, position='after_iter'):
        super().__init__(name, priority, position)
        self._fn = fn

    def __call__(self, engine):
        self._fn(engine)


class LoadCkptHook(Hook):

    def __init__(self, load_path: str, **kwargs):
        super().__init__('load_ckpt', 0, 'before_run')
        self._load_path = load_path

    def __call__(self, engine: 'BaseLearner') -> None:
        if not self._load_path:
            return
        state = read_file(self._load_path)
        engine.policy.load_state_dict(state)
        if 'last_iter' in state:
            engine.last_iter.update(state['last_iter'])
        engine.info(f'load checkpoint from {self._load_path}')


class SaveCkptHook(Hook):

    def __init__(self, train_freq: int = 100, **kwargs):
        super().__init__('save_ckpt', 100, 'after_iter')
        self._freq = train_freq

    def __call__(self, engine: 'BaseLearner') -> None:
        if engine.last_iter.val == 0 or engine.last_iter.val % self._freq != 0:
            return
        engine.save_checkpoint(f'iteration_{engine.last_iter.val}.pth.tar')


class LogShowHook(Hook):

    def __init__(self, freq: int = 100, **kwargs):
        super().__init__('log_show', 200, 'after_iter')
        self._freq = freq

    def __call__(self, engine: 'BaseLearner') -> None:
        if engine.rank != 0 or engine.last_iter.val % self._freq != 0:
            return
        info = {k: np.mean(v) for k, v in engine.log_buffer.items() if len(v) > 0}
        engine.log_buffer.clear()
        scalars = {k: v for k, v in info.items() if np.isscalar(v)}
        engine.info(f"iter {engine.last_iter.val}: " + ", ".join(f"{k}={v:.4g}" for k, v in scalars.items()))
        if engine.tb_logger is not None:
            for k, v in scalars.items():
                engine.tb_logger.add_scalar(f'learner_iter/{k}', v, engine.last_iter.val)


def build_learner_hook_by_cfg(cfg: EasyDict) -> Dict[str, List[Hook]]:
    hooks = {'before_run': [], 'before_iter': [], 'after_iter': [], 'after_run': []}
    if cfg.get('load_ckpt_before_run'):
        hooks['before_run'].append(LoadCkptHook(cfg.load_ckpt_before_run))
    hooks['after_iter'].append(LogShowHook(cfg.get('log_show_after_iter', 100)))
    if cfg.get('save_ckpt_after_iter', None):
        hooks['after_iter'].append(SaveCkptHook(cfg.save_ckpt_after_iter))
    if cfg.get('save_ckpt_after_run', True):
        hooks['after_run'].append(
            LambdaHook('save_final', lambda e: e.save_checkpoint('final.pth.tar'), position='after_run')
        )
    return hooks


register_learner_hook = None  # placeholder for API parity; use add_hook


class CountVar:

    def __init__(self, v=0):
        self._v = v

    @property
    def val(self):
        return self._v

    def update(self, v):
        self._v = v

    def add(self, n):
        self._v += n


@LEARNER_REGISTRY.register('base')
class BaseLearner:

    config = dict(
        train_iterations=int(1e9),
        dataloader=dict(num_workers=0, ),
        log_policy=True,
        hook=dict(
            load_ckpt_before_run='',
            log_show_after_iter=100,
            save_ckpt_after_iter=10000,
            save_ckpt_after_run=True,
        ),
    )

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))

    def __init__(
        self,
        cfg: EasyDict,
        policy=None,
        tb_logger=None,
        dist_info=None,
        exp_name: str = 'default_experiment',
        instance_name: str = 'learner',
    ):
        self._cfg = deep_merge_dicts(self.default_config(), cfg or EasyDict({}))
        self._exp_name = exp_name
        self._instance_name = instance_name
        self._ckpt_dir = os.path.join(exp_name, 'ckpt')
        self._timer = EasyTimer(cuda=False)
        if dist_info is not None:
            self._rank, self._world_size = dist_info
        else:
            self._rank, self._world_size = get_rank(), 1
        self._logger, self._tb_logger = build_logger(
            os.path.join(exp_name, 'log', instance_name), instance_name, need_tb=(tb_logger is None and self._rank == 0)
        )
        if tb_logger is not None:
            self._tb_logger = tb_logger
        self._end_flag = False
        self.last_iter = CountVar(0)
        self.log_buffer: Dict[str, list] = {}
        self._hooks = build_learner_hook_by_cfg(self._cfg.hook)
        self._collector_envstep = 0
        if policy is not None:
            self.policy = policy
        else:
            self._policy = None
        self.priority_info = {}

    # -------------------------------------------------------------- props
    @property
    def policy(self):
        return self._policy

    @policy.setter
    def policy(self, policy):
        self._policy = policy

    @property
    def rank(self):
        return self._rank

    @property
    def tb_logger(self):
        return self._tb_logger

    @property
    def train_iter(self) -> int:
        return self.last_iter.val

    @property
    def monitor(self):
        return None

    @property
    def learn_info(self) -> dict:
        return {'learner_step': self.last_iter.val, 'priority_info': self.priority_info, 'learner_done': False}

    def info(self, msg: str):
        if self._logger:
            self._logger.info(f'[{self._instance_name}] {msg}')

    def debug(self, msg: str):
        if self._logger:
            self._logger.debug(msg)

    # -------------------------------------------------------------- hooks
    def add_hook(self, hook: Hook):
        self._hooks[hook.position].append(hook)
        self._hooks[hook.position].sort(key=lambda h: h.priority)

    def call_hook(self, name: str) -> None:
        for hook in self._hooks.get(name, []):
            hook(self)

    # --------------------------------------------------------------- train
    def train(self, data: Union[dict, List[dict]], envstep: int = -1, policy_kwargs: Optional[dict] = None) -> None:
        assert self._policy is not None
        self.call_hook('before_iter')
        if envstep >= 0:
            self._collector_envstep = envstep
        with self._timer:
            log_vars = self._policy.forward(data, **(policy_kwargs or {}))
        if isinstance(log_vars, dict):
            log_vars_list = [log_vars]
        else:
            log_vars_list = list(log_vars)
        for log in log_vars_list:
            priority = log.pop('priority', None)
            if priority is not None and isinstance(data, list) and len(data) > 0 and isinstance(data[0], dict) \
                    and 'replay_buffer_idx' in data[0]:
                self.priority_info = {
                    'priority': priority,
                    'replay_unique_id': [d.get('replay_unique_id') for d in data],
                    'replay_buffer_idx': [d.get('replay_buffer_idx') for d in data],
                }
            elif priority is not None:
                self.priority_info = {'priority': priority}
            for k, v in log.items():
                if np.isscalar(v):
                    self.log_buffer.setdefault(k, []).append(v)
            self.log_buffer.setdefault('train_time', []).append(self._timer.value)
            self.last_iter.add(1)
        self.call_hook('after_iter')

    def start(self) -> None:
        self.call_hook('before_run')

    def save_checkpoint(self, ckpt_name: Optional[str] = None) -> None:
        if self._rank != 0:
            return
        ckpt_name = ckpt_name or f'iteration_{self.last_iter.val}.pth.tar'
        path = os.path.join(self._ckpt_dir, ckpt_name)
        state = self._policy.state_dict()
        state['last_iter'] = self.last_iter.val
        save_file(path, state)
        self.info(f'save checkpoint to {path}')

    def close(self) -> None:
        if self._end_flag:
            return
        self._end_flag = True
        self.call_hook('after_run')
        if self._tb_logger is not None:
            self._tb_logger.flush()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def create_learner(cfg: EasyDict, **kwargs) -> BaseLearner:
    return LEARNER_REGISTRY.build(cfg.get('type', 'base'), cfg=cfg, **kwargs)
