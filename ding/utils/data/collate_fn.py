"""Batch collation/decollation for transition dicts.

Parity: reference ding/utils/data/collate_fn.py (default_collate:80,
timestep_collate:172, diff_shape_collate:226, default_decollate:293).
"""
from collections.abc import Mapping, Sequence
from typing import Any, Dict, List, Union

import numpy as np
import torch


def ttorch_collate(x):
    return default_collate(x)


def default_collate(batch: Sequence, cat_1dim: bool = True, ignore_prefix: list = ('collate_ignore', )) -> Any:
    """Stack a list of samples: tensors -> stacked tensor (1-elem tensors are
    concatenated to [B] when cat_1dim), scalars -> tensor, dicts -> dict of
    collated values, namedtuple/sequence -> recursively collated."""
    if len(batch) == 0:
        return batch
    elem = batch[0]
    if isinstance(elem, torch.Tensor):
        if elem.shape == (1, ) and cat_1dim:
            # tolerate mixed 0-dim / [1] entries (random-collect vs policy
            # transitions can disagree on scalar action shape)
            return torch.cat([b.reshape(1) for b in batch], 0)
        return torch.stack(batch, 0)
    if isinstance(elem, np.ndarray):
        return default_collate([torch.as_tensor(b) for b in batch], cat_1dim=cat_1dim)
    if isinstance(elem, (np.floating, float)):
        return torch.tensor(batch, dtype=torch.float32)
    if isinstance(elem, (np.integer, int)):
        return torch.tensor(batch)
    if isinstance(elem, np.bool_):
        return torch.tensor(batch)
    if isinstance(elem, (str, bytes)):
        return list(batch)
    if isinstance(elem, bool):
        return torch.tensor(batch)
    if isinstance(elem, tuple) and hasattr(elem, '_fields'):
        return type(elem)(*(default_collate(s, cat_1dim=cat_1dim) for s in zip(*batch)))
    if isinstance(elem, Mapping):
        out = {}
        for key in elem:
            if any(str(key).startswith(p) for p in ignore_prefix):
                out[key] = [d[key] for d in batch]
            elif elem[key] is None:
                out[key] = None
            else:
                out[key] = default_collate([d[key] for d in batch], cat_1dim=cat_1dim)
        return out
    if isinstance(elem, Sequence):
        # keep per-sample lists (e.g. prev_state) as transposed lists
        if isinstance(elem[0], (dict, type(None))) or (isinstance(elem[0], torch.Tensor) and elem[0].dim() >= 1):
            try:
                transposed = list(zip(*batch))
                return [default_collate(list(s), cat_1dim=cat_1dim) for s in transposed]
            except Exception:
                return list(batch)
        return default_collate([torch.as_tensor(b) for b in batch], cat_1dim=cat_1dim)
    if elem is None:
        return batch
    raise TypeError(f"default_collate: unsupported type {type(elem)}")


def timestep_collate(batch: List[Dict[str, Any]]) -> Dict[str, Any]:
    """Collate unrolled samples: each sample is a dict of per-timestep lists;
    output tensors are [T, B, ...]; 'prev_state' kept as list of per-step
    per-sample states."""
    elem = batch[0]
    prev_state = None
    if 'prev_state' in elem:
        # [B][T] -> [T][B]. Do NOT pop in place: buffers hand out references
        # to stored samples, and a destructive pop breaks re-sampling the
        # same transition (update_per_collect > buffer turnover).
        prev_state = list(zip(*[d['prev_state'] for d in batch]))
        batch = [{k: v for k, v in d.items() if k != 'prev_state'} for d in batch]
        elem = batch[0]
    def _stack_time(vals):
        """vals: B samples, each a T-list (or [T,...] tensor) -> [T, B, ...]."""
        if isinstance(vals[0], dict):
            return {k: _stack_time([v[k] for v in vals]) for k in vals[0]}
        if isinstance(vals[0], (list, tuple)):
            stacked = [default_collate(list(ts_vals)) for ts_vals in zip(*vals)]  # per timestep over batch
            if isinstance(stacked[0], torch.Tensor):
                return torch.stack(stacked, 0)
            if isinstance(stacked[0], dict):
                return {k: torch.stack([s[k] for s in stacked], 0) for k in stacked[0]}
            return stacked
        if isinstance(vals[0], torch.Tensor) and vals[0].dim() >= 1:
            return torch.stack(vals, 1)  # already [T, ...] per sample
        return default_collate(vals)

    out = {}
    for key in elem:
        vals = [d[key] for d in batch]
        # an optional per-sample field (e.g. weight=None) collapses to None
        if all(v is None for v in vals) or (
            isinstance(vals[0], (list, tuple)) and all(x is None for v in vals for x in v)
        ):
            out[key] = None
            continue
        out[key] = _stack_time(vals)
    if prev_state is not None:
        out['prev_state'] = [list(s) for s in prev_state]
    return out


def diff_shape_collate(batch: Sequence) -> Any:
    """Like default_collate but tolerates per-sample shape differences by
    returning lists where stacking fails."""
    elem = batch[0]
    if isinstance(elem, torch.Tensor):
        shapes = set(tuple(b.shape) for b in batch)
        if len(shapes) != 1:
            return list(batch)
        return default_collate(batch)
    if isinstance(elem, np.ndarray):
        return diff_shape_collate([torch.as_tensor(b) for b in batch])
    if isinstance(elem, Mapping):
        return {k: diff_shape_collate([d[k] for d in batch]) for k in elem}
    if isinstance(elem, Sequence) and not isinstance(elem, (str, bytes)):
        return [diff_shape_collate(list(s)) for s in zip(*batch)]
    return default_collate(batch)


def default_decollate(batch: Union[torch.Tensor, Sequence, Mapping], ignore: List[str] = ['prev_state', 'prev_actor_state', 'prev_critic_state']) -> List[Any]:
    """Split a batch back into per-sample structures (inverse of collate)."""
    if isinstance(batch, torch.Tensor):
        return list(torch.unbind(batch, dim=0))
    if isinstance(batch, Mapping):
        keys = list(batch.keys())
        parts = {}
        length = None
        for k in keys:
            if k in ignore:
                continue
            parts[k] = default_decollate(batch[k], ignore)
            if length is None and isinstance(parts[k], list):
                length = len(parts[k])
        length = length or 1
        out = []
        for i in range(length):
            d = {}
            for k in keys:
                if k in ignore:
                    d[k] = batch[k][i] if isinstance(batch[k], (list, tuple)) and len(batch[k]) == length else batch[k]
                else:
                    d[k] = parts[k][i] if isinstance(parts[k], list) and len(parts[k]) == length else batch[k]
            out.append(d)
        return out
    if isinstance(batch, Sequence) and not isinstance(batch, (str, bytes)):
        elem_lists = [default_decollate(b, ignore) for b in batch]
        if all(isinstance(e, list) for e in elem_lists):
            length = len(elem_lists[0])
            return [[e[i] for e in elem_lists] for i in range(length)]
        return list(batch)
    return batch
