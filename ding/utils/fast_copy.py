"""Type-dispatched fast deep copy for transition payloads.

Parity: reference ding/utils/fast_copy.py:6 — avoids copy.deepcopy's
reflection cost on the collector hot path.
"""
from typing import Any, List

import numpy as np
import torch


def _copy_tensor(x: torch.Tensor) -> torch.Tensor:
    return x.clone()


def _copy_ndarray(x: np.ndarray) -> np.ndarray:
    return np.copy(x)


def _copy_dict(x: dict) -> dict:
    return {k: fast_copy(v) for k, v in x.items()}


def _copy_list(x: List) -> List:
    return [fast_copy(v) for v in x]


_DISPATCH = {torch.Tensor: _copy_tensor, np.ndarray: _copy_ndarray, dict: _copy_dict, list: _copy_list}


def fast_copy(x: Any) -> Any:
    t = type(x)
    fn = _DISPATCH.get(t)
    if fn is not None:
        return fn(x)
    if isinstance(x, (int, float, str, bool, bytes, type(None))):
        return x
    import copy
    return copy.deepcopy(x)
