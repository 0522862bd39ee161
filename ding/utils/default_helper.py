"""General helpers: seeding, dict/list reshaping, merge, squeeze.

Parity: reference ding/utils/default_helper.py (lists_to_dicts, dicts_to_lists,
set_pkg_seed, deep_merge_dicts, squeeze, one_time_warning, ...).
"""
import copy
import logging
import random
from functools import lru_cache
from typing import Any, Dict, List, Optional, Sequence, Union

import numpy as np
import torch

from .edict import EasyDict


def set_pkg_seed(seed: int, use_cuda: bool = True) -> None:
    """Seed python/numpy/torch (+ HIP device RNG when available)."""
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if use_cuda and torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def lists_to_dicts(data: Sequence[Union[dict, tuple]], recursive: bool = False) -> Union[dict, tuple]:
    """[{k: v1}, {k: v2}] -> {k: [v1, v2]}"""
    if len(data) == 0:
        raise ValueError("empty data")
    first = data[0]
    if isinstance(first, dict):
        keys = first.keys()
        _no_recurse = ('prev_state', 'prev_actor_state', 'prev_critic_state')
        out = {}
        for k in keys:
            vals = [d[k] for d in data]
            if recursive and isinstance(vals[0], dict) and k not in _no_recurse:
                vals = lists_to_dicts(vals, recursive=True)
            out[k] = vals
        if isinstance(first, EasyDict):
            out = EasyDict(out)
        return out
    elif isinstance(first, tuple) and hasattr(first, "_fields"):  # namedtuple
        return type(first)(*[[getattr(d, f) for d in data] for f in first._fields])
    else:
        raise TypeError(f"unsupported element type: {type(first)}")


def dicts_to_lists(data: Dict[str, List[Any]]) -> List[Dict[str, Any]]:
    """{k: [v1, v2]} -> [{k: v1}, {k: v2}]"""
    if len(data) == 0:
        raise ValueError("empty data")
    n = len(next(iter(data.values())))
    return [{k: v[i] for k, v in data.items()} for i in range(n)]


def deep_merge_dicts(original: dict, new_dict: dict) -> dict:
    """Return a new dict: ``new_dict`` values override ``original`` recursively."""
    out = copy.deepcopy(original)
    _deep_update(out, new_dict)
    return out


def _deep_update(target: dict, src: dict) -> dict:
    for k, v in src.items():
        if k in target and isinstance(target[k], dict) and isinstance(v, dict):
            _deep_update(target[k], v)
        else:
            target[k] = copy.deepcopy(v)
    return target


def deep_update(original: dict, new_dict: dict, new_keys_allowed: bool = True, whitelist=None) -> dict:
    """In-place recursive update (reference deep_update semantics, simplified)."""
    return _deep_update(original, new_dict)


def squeeze(data: Any) -> Any:
    """Unwrap single-element tuples/lists; pass through scalars."""
    if isinstance(data, (tuple, list)):
        if len(data) == 1:
            return data[0]
        return tuple(data)
    if isinstance(data, dict):
        if len(data) == 1:
            return next(iter(data.values()))
    return data


@lru_cache(maxsize=None)
def one_time_warning(msg: str) -> None:
    logging.getLogger("ding").warning(msg)


def error_wrapper(fn, default_ret, warning_msg: str = ""):
    """Call fn(); on exception return ``default_ret`` (optionally warn once)."""

    def wrapper(*args, **kwargs):
        try:
            return fn(*args, **kwargs)
        except Exception:
            if warning_msg:
                one_time_warning(warning_msg)
            return default_ret

    return wrapper


class LimitedSpaceContainer:
    """A counter with [min, max] occupancy semantics (used by coordinator)."""

    def __init__(self, min_val: int, max_val: int):
        self.min_val = min_val
        self.max_val = max_val
        self.cur = min_val

    def get_residual_space(self) -> int:
        ret = self.max_val - self.cur
        self.cur = self.max_val
        return ret

    def acquire_space(self) -> bool:
        if self.cur < self.max_val:
            self.cur += 1
            return True
        return False

    def release_space(self) -> None:
        self.cur = max(self.min_val, self.cur - 1)

    def increase_space(self) -> None:
        self.max_val += 1

    def decrease_space(self) -> None:
        self.max_val = max(self.min_val, self.max_val - 1)


def get_shape0(data: Any) -> int:
    """Leading dimension of (possibly nested) tensor data."""
    if isinstance(data, torch.Tensor):
        return data.shape[0]
    if isinstance(data, dict):
        return get_shape0(next(iter(data.values())))
    if isinstance(data, (list, tuple)):
        return get_shape0(data[0])
    raise TypeError(type(data))


def split_data_generator(data: dict, split_size: int, shuffle: bool = True):
    """Yield minibatch dicts of ``split_size`` rows from a dict of tensors.

    Parity: ding/utils/default_helper.py split_data_generator used by PPO
    epoch training (policy/ppo.py).
    """
    length = None
    for v in data.values():
        if isinstance(v, torch.Tensor) and v.dim() > 0:
            length = v.shape[0]
            break
    assert length is not None, "no tensor field to infer batch size"
    if shuffle:
        indices = torch.randperm(length, device="cpu")
    else:
        indices = torch.arange(length)
    def _index(v, idx):
        if isinstance(v, torch.Tensor) and v.dim() > 0 and v.shape[0] == length:
            return v[idx.to(v.device)]
        if isinstance(v, dict):
            return {k: _index(sub, idx) for k, sub in v.items()}
        return v

    for start in range(0, length - split_size + 1, split_size):
        idx = indices[start:start + split_size]
        yield {k: _index(v, idx) for k, v in data.items()}


def flatten_dict(data: dict, delimiter: str = "/", prefix: str = "") -> dict:
    out = {}
    for k, v in data.items():
        key = f"{prefix}{delimiter}{k}" if prefix else str(k)
        if isinstance(v, dict):
            out.update(flatten_dict(v, delimiter, key))
        else:
            out[key] = v
    return out


class RunningMeanStd:
    """Streaming mean/std via parallel-variance merge (Chan et al.).

    Parity: reference env_wrappers.py:629 RunningMeanStd (obs/reward norm).
    """

    def __init__(self, epsilon: float = 1e-4, shape=(), device=None):
        self._eps = epsilon
        self._shape = shape
        self.reset()

    def reset(self):
        self.mean = np.zeros(self._shape, dtype=np.float64)
        self.var = np.ones(self._shape, dtype=np.float64)
        self.count = self._eps

    def update(self, x: np.ndarray):
        x = np.asarray(x, dtype=np.float64)
        batch_mean = x.mean(axis=0)
        batch_var = x.var(axis=0)
        batch_count = x.shape[0]
        delta = batch_mean - self.mean
        tot = self.count + batch_count
        self.mean = self.mean + delta * batch_count / tot
        m_a = self.var * self.count
        m_b = batch_var * batch_count
        m2 = m_a + m_b + delta ** 2 * self.count * batch_count / tot
        self.var = m2 / tot
        self.count = tot

    @property
    def std(self):
        return np.sqrt(self.var + 1e-8)


def get_task_uid() -> str:
    """Short unique id for parallel-pipeline task names."""
    import uuid
    return uuid.uuid4().hex[:8]
