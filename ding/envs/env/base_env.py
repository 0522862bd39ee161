"""Env ABC and timestep container.

Parity: reference ding/envs/env/base_env.py (BaseEnv:17, BaseEnvTimestep:88,
get_vec_env_setting:141, get_env_cls, create_model_env).
"""
import copy
from abc import ABC, abstractmethod
from collections import namedtuple
from typing import Any, Callable, List, Optional, Tuple

from ding.utils import ENV_REGISTRY, EasyDict, import_module

BaseEnvTimestep = namedtuple('BaseEnvTimestep', ['obs', 'reward', 'done', 'info'])


class BaseEnv(ABC):
    """Decision-intelligence env interface (gym-compatible step/reset/seed/
    close, plus static config splitting for collector/evaluator variants)."""

    @abstractmethod
    def __init__(self, cfg: dict) -> None:
        raise NotImplementedError

    @abstractmethod
    def reset(self) -> Any:
        raise NotImplementedError

    @abstractmethod
    def close(self) -> None:
        raise NotImplementedError

    @abstractmethod
    def step(self, action: Any) -> BaseEnvTimestep:
        raise NotImplementedError

    @abstractmethod
    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        raise NotImplementedError

    @abstractmethod
    def __repr__(self) -> str:
        raise NotImplementedError

    @staticmethod
    def create_collector_env_cfg(cfg: dict) -> List[dict]:
        collector_env_num = cfg.pop('collector_env_num', 1)
        cfg = copy.deepcopy(cfg)
        if 'is_train' in cfg:
            cfg.is_train = True
        return [cfg for _ in range(collector_env_num)]

    @staticmethod
    def create_evaluator_env_cfg(cfg: dict) -> List[dict]:
        evaluator_env_num = cfg.pop('evaluator_env_num', 1)
        cfg = copy.deepcopy(cfg)
        if 'is_train' in cfg:
            cfg.is_train = False
        return [cfg for _ in range(evaluator_env_num)]

    def enable_save_replay(self, replay_path: str) -> None:
        raise NotImplementedError

    def random_action(self) -> Any:
        return self.action_space.sample()


def get_env_cls(cfg: EasyDict) -> type:
    import_module(cfg.get('import_names', []))
    return ENV_REGISTRY.get(cfg.type)


def get_vec_env_setting(cfg: EasyDict, collect: bool = True, eval_: bool = True) -> Tuple[type, List[dict], List[dict]]:
    """Return (env_cls, collector_env_cfgs, evaluator_env_cfgs)."""
    import_module(cfg.get('import_names', []))
    env_cls = ENV_REGISTRY.get(cfg.type)
    collector_env_cfg = env_cls.create_collector_env_cfg(cfg) if collect else None
    evaluator_env_cfg = env_cls.create_evaluator_env_cfg(cfg) if eval_ else None
    return env_cls, collector_env_cfg, evaluator_env_cfg


def create_env(cfg: EasyDict) -> BaseEnv:
    import_module(cfg.get('import_names', []))
    return ENV_REGISTRY.build(cfg.type, cfg=cfg)


def create_model_env(cfg: EasyDict):
    """Build a model-env (world-model rollout env for mb-RL) from its registry
    entry; remaining cfg keys go to the constructor (reference base_env.py:176)."""
    import copy as _copy
    cfg = _copy.deepcopy(cfg)
    env_fn = get_env_cls(cfg)
    cfg.pop('import_names', None)
    cfg.pop('type', None)
    return env_fn(**cfg)
