"""Buffer middleware: PER, clone, use-count, staleness, sample-range view,
padding, group sampling.

Parity: reference ding/data/buffer/middleware/*.py. The PER middleware keeps
its priority state in the host C++ segment tree (ding.utils._ctree) — the
reference's numba version replaced per SURVEY §2.9b.
"""
import copy
from typing import Any, Callable, List, Optional

import numpy as np

from ding.utils import SumSegmentTree, MinSegmentTree
from .buffer import BufferedData


class PriorityExperienceReplay:
    """Proportional PER over an index-addressed buffer."""

    def __init__(
        self,
        buffer_,
        IS_weight: bool = True,
        priority_power_factor: float = 0.6,
        IS_weight_power_factor: float = 0.4,
        IS_weight_anneal_train_iter: int = int(1e5),
    ):
        self.buffer = buffer_
        self.buffer_idx = {}
        self.buffer_size = buffer_.size
        self.IS_weight = IS_weight
        self.priority_power_factor = priority_power_factor
        self.IS_weight_power_factor = IS_weight_power_factor
        self.IS_weight_anneal_train_iter = IS_weight_anneal_train_iter
        capacity = int(np.power(2, np.ceil(np.log2(self.buffer_size))))
        self.sum_tree = SumSegmentTree(capacity)
        if self.IS_weight:
            self.min_tree = MinSegmentTree(capacity)
        self.delta_anneal = (1 - self.IS_weight_power_factor) / self.IS_weight_anneal_train_iter
        self.pivot = 0
        self.max_priority = 1.0

    def push(self, chain: Callable, data: Any, *args, **kwargs) -> BufferedData:
        if 'meta' in kwargs and kwargs['meta'] is not None:
            meta = kwargs['meta']
        else:
            meta = {}
            kwargs['meta'] = meta
        priority = meta.get('priority', self.max_priority)
        buffered = chain(data, *args, **kwargs)
        index = buffered.index
        weight = priority ** self.priority_power_factor
        self.sum_tree[self.pivot] = weight
        if self.IS_weight:
            self.min_tree[self.pivot] = weight
        buffered.meta['priority'] = priority
        buffered.meta['priority_idx'] = self.pivot
        self.buffer_idx[self.pivot] = index
        self.pivot = (self.pivot + 1) % self.buffer_size
        return buffered

    def sample(self, chain: Callable, size: Optional[int] = None, *args, **kwargs):
        if kwargs.get('indices') is not None or size is None:
            return chain(size, *args, **kwargs)
        # stratified prefix-sum sampling through the C++ tree
        total = self.sum_tree.reduce()
        if total <= 0:
            return chain(size, *args, **kwargs)
        mass = (np.random.rand(size) + np.arange(size)) / size * total
        idx = self.sum_tree.find_prefixsum_idx(mass)
        indices = [self.buffer_idx[int(i)] for i in idx]
        kwargs['indices'] = indices
        data = chain(None, *args, **kwargs)
        if self.IS_weight:
            n = self.buffer.count()
            p_min = self.min_tree.reduce() / total
            max_weight = (n * p_min) ** (-self.IS_weight_power_factor)
            for bd, i in zip(data, idx):
                p = self.sum_tree[int(i)] / total
                bd.meta['priority_IS'] = float(((n * p) ** (-self.IS_weight_power_factor)) / max_weight)
            self.IS_weight_power_factor = min(1.0, self.IS_weight_power_factor + self.delta_anneal)
        return data

    def update(self, chain: Callable, index: str, data: Optional[Any] = None, meta: Optional[dict] = None, *args,
               **kwargs) -> bool:
        update_flag = chain(index, data, meta, *args, **kwargs)
        if update_flag and meta is not None and 'priority' in meta:
            try:
                bd = self.buffer.sample(indices=[index])[0]
            except KeyError:
                return update_flag
            pidx = bd.meta.get('priority_idx')
            if pidx is not None:
                new_p = max(meta['priority'], 1e-6)
                weight = new_p ** self.priority_power_factor
                self.sum_tree[pidx] = weight
                if self.IS_weight:
                    self.min_tree[pidx] = weight
                self.max_priority = max(self.max_priority, new_p)
                bd.meta['priority'] = new_p
        return update_flag

    def delete(self, chain: Callable, index: str, *args, **kwargs):
        return chain(index, *args, **kwargs)

    def clear(self, chain: Callable, *args, **kwargs):
        self.max_priority = 1.0
        capacity = int(np.power(2, np.ceil(np.log2(self.buffer_size))))
        self.sum_tree = SumSegmentTree(capacity)
        if self.IS_weight:
            self.min_tree = MinSegmentTree(capacity)
        self.buffer_idx = {}
        self.pivot = 0
        return chain(*args, **kwargs)

    def state_dict(self) -> dict:
        return {
            'max_priority': self.max_priority,
            'IS_weight_power_factor': self.IS_weight_power_factor,
            'pivot': self.pivot,
            'sum_tree_value': self.sum_tree.value.copy(),
            'min_tree_value': self.min_tree.value.copy() if self.IS_weight else None,
            'buffer_idx': dict(self.buffer_idx),
        }

    def load_state_dict(self, d: dict):
        self.max_priority = d['max_priority']
        self.IS_weight_power_factor = d['IS_weight_power_factor']
        self.pivot = d['pivot']
        self.sum_tree.value[:] = d['sum_tree_value']
        if self.IS_weight and d['min_tree_value'] is not None:
            self.min_tree.value[:] = d['min_tree_value']
        self.buffer_idx = d['buffer_idx']

    def __call__(self, action: str, chain: Callable, *args, **kwargs) -> Any:
        if action in ('push', 'sample', 'update', 'delete', 'clear'):
            return getattr(self, action)(chain, *args, **kwargs)
        return chain(*args, **kwargs)


def clone_object():
    """Deep-copy data on push and sample (isolate buffer from callers)."""
    from ding.utils import fast_copy

    def _clone(action: str, chain: Callable, *args, **kwargs):
        if action == 'push':
            args = [fast_copy(args[0]), *args[1:]] if args else args
            return chain(*args, **kwargs)
        if action == 'sample':
            data = chain(*args, **kwargs)
            return [BufferedData(data=fast_copy(d.data), index=d.index, meta=fast_copy(d.meta)) for d in data]
        return chain(*args, **kwargs)

    return _clone


def use_time_check(buffer_, max_use: int = float("inf")):
    """Delete items sampled more than max_use times."""

    def _use_time(action: str, chain: Callable, *args, **kwargs):
        if action == 'sample':
            data = chain(*args, **kwargs)
            to_delete = []
            for item in data:
                items = item if isinstance(item, list) else [item]
                for d in items:
                    d.meta['use_count'] = d.meta.get('use_count', 0) + 1
                    if d.meta['use_count'] >= max_use:
                        to_delete.append(d.index)
            if to_delete:
                buffer_.delete(to_delete)
            return data
        return chain(*args, **kwargs)

    return _use_time


def staleness_check(buffer_, max_staleness: int = float("inf")):
    """On sample(train_iter=...), drop items whose train_iter gap exceeds
    max_staleness, then resample."""

    def _staleness(action: str, chain: Callable, *args, **kwargs):
        if action == 'push':
            meta = kwargs.get('meta') or (args[1] if len(args) > 1 else None)
            assert meta is not None and 'train_iter_data_collected' in meta, \
                "staleness_check requires meta['train_iter_data_collected'] on push"
            return chain(*args, **kwargs)
        if action == 'sample':
            train_iter_now = kwargs.pop('train_iter_sample_data', None)
            if train_iter_now is not None:
                stale = [
                    bd.index for bd in buffer_.storage
                    if train_iter_now - bd.meta.get('train_iter_data_collected', train_iter_now) > max_staleness
                ]
                if stale:
                    buffer_.delete(stale)
            return chain(*args, **kwargs)
        return chain(*args, **kwargs)

    return _staleness


def sample_range_view(buffer_, start: Optional[int] = None, end: Optional[int] = None):
    """Restrict sampling to a slice of the storage (e.g. most recent N)."""
    s = slice(start, end)

    def _range(action: str, chain: Callable, *args, **kwargs):
        if action == 'sample':
            kwargs['sample_range'] = s
            return chain(*args, **kwargs)
        return chain(*args, **kwargs)

    return _range


def padding(policy: str = "none"):
    """Pad grouped samples to equal length ('none': repeat last item)."""

    def _padding(action: str, chain: Callable, *args, **kwargs):
        if action == 'sample':
            data = chain(*args, **kwargs)
            if data and isinstance(data[0], list):
                max_len = max(len(ep) for ep in data)
                for ep in data:
                    while len(ep) < max_len:
                        ep.append(copy.deepcopy(ep[-1]))
            return data
        return chain(*args, **kwargs)

    return _padding


def group_sample(size_in_group: int, ordered_in_group: bool = True, max_use_in_group: bool = True):
    """Subsample fixed-size (optionally contiguous) windows from grouped
    episodes (R2D2 sequence sampling)."""
    import random

    def _group(action: str, chain: Callable, *args, **kwargs):
        if action == 'sample':
            data = chain(*args, **kwargs)
            out = []
            for ep in data:
                if not isinstance(ep, list):
                    out.append(ep)
                    continue
                if len(ep) <= size_in_group:
                    out.append(ep)
                elif ordered_in_group:
                    start = random.randint(0, len(ep) - size_in_group)
                    out.append(ep[start:start + size_in_group])
                else:
                    out.append(random.sample(ep, size_in_group))
            return out
        return chain(*args, **kwargs)

    return _group
