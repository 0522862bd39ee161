"""Deque-backed replay buffer with O(1) index lookup and grouped sampling.

Parity: reference ding/data/buffer/deque_buffer.py (DequeBuffer:51,
BufferIndex:15).
"""
import itertools
import random
import uuid
from collections import defaultdict, deque
from typing import Any, Iterable, List, Optional, Union

from ding.utils import BUFFER_REGISTRY
from .buffer import Buffer, BufferedData, apply_middleware


class BufferIndex:
    """index-string -> deque position map that survives left-eviction by
    tracking a monotonically increasing offset."""

    def __init__(self, maxlen: int):
        self.maxlen = maxlen
        self._map = {}
        self._head = 0  # count of evicted items
        self._tail = 0  # total pushed

    def append(self, key: str):
        self._map[key] = self._tail
        self._tail += 1
        if self._tail - self._head > self.maxlen:
            self._head += 1

    def evict_below(self, head: int):
        self._head = max(self._head, head)

    def get(self, key: str) -> Optional[int]:
        pos = self._map.get(key)
        if pos is None or pos < self._head:
            return None
        return pos - self._head

    def remove(self, key: str):
        self._map.pop(key, None)

    def clear(self):
        self._map.clear()
        self._head = self._tail = 0

    def gc(self):
        if len(self._map) > 2 * self.maxlen:
            self._map = {k: v for k, v in self._map.items() if v >= self._head}


@BUFFER_REGISTRY.register('deque')
class DequeBuffer(Buffer):

    def __init__(self, size: int, sliced: bool = False):
        super().__init__(size=size)
        self.storage: deque = deque(maxlen=size)
        self.indices = BufferIndex(maxlen=size)
        self.sliced = sliced
        self._pushed = 0

    @apply_middleware("push")
    def push(self, data: Any, meta: Optional[dict] = None) -> BufferedData:
        return self._push(data, meta)

    def _push(self, data: Any, meta: Optional[dict] = None) -> BufferedData:
        index = uuid.uuid4().hex
        bd = BufferedData(data=data, index=index, meta=meta or {})
        self.storage.append(bd)
        self.indices.append(index)
        self._pushed += 1
        self.indices.evict_below(self._pushed - len(self.storage))
        self.indices.gc()
        return bd

    @apply_middleware("sample")
    def sample(
        self,
        size: Optional[int] = None,
        indices: Optional[List[str]] = None,
        replace: bool = False,
        sample_range: Optional[slice] = None,
        ignore_insufficient: bool = False,
        groupby: Optional[str] = None,
        unroll_len: Optional[int] = None,
    ) -> Union[List[BufferedData], List[List[BufferedData]]]:
        storage = self.storage
        if sample_range is not None:
            storage = list(itertools.islice(storage, *sample_range.indices(len(storage))))
        if indices is not None:
            pos = [self.indices.get(i) for i in indices]
            missing = [i for i, p in zip(indices, pos) if p is None]
            if missing:
                raise KeyError(f"indices not in buffer: {missing[:5]}")
            return [self.storage[p] for p in pos]
        if groupby is not None:
            return self._sample_by_group(size, groupby, unroll_len, storage, replace)
        value_error = None
        sampled = []
        if size is None:
            raise ValueError("either size or indices must be provided")
        if len(storage) < size:
            if ignore_insufficient:
                size = len(storage)
            else:
                value_error = ValueError(
                    f"buffer has {len(storage)} items but {size} requested (set ignore_insufficient to allow)"
                )
        if value_error:
            raise value_error
        if replace:
            sampled = random.choices(list(storage), k=size)
        else:
            sampled = random.sample(list(storage), k=size)
        return sampled

    def _sample_by_group(self, size, groupby, unroll_len, storage, replace) -> List[List[BufferedData]]:
        groups = defaultdict(list)
        for bd in storage:
            key = bd.meta.get(groupby)
            groups[key].append(bd)
        keys = list(groups.keys())
        if size is not None and len(keys) < size and not replace:
            raise ValueError(f"only {len(keys)} groups for requested {size}")
        chosen = random.sample(keys, k=size) if not replace else random.choices(keys, k=size)
        out = []
        for k in chosen:
            episode = groups[k]
            if unroll_len is not None and len(episode) > unroll_len:
                start = random.randint(0, len(episode) - unroll_len)
                episode = episode[start:start + unroll_len]
            out.append(episode)
        return out

    @apply_middleware("update")
    def update(self, index: str, data: Optional[Any] = None, meta: Optional[dict] = None) -> bool:
        pos = self.indices.get(index)
        if pos is None:
            return False
        bd = self.storage[pos]
        if data is not None:
            bd.data = data
        if meta is not None:
            bd.meta = meta
        return True

    @apply_middleware("delete")
    def delete(self, indices: Union[str, Iterable[str]]) -> None:
        if isinstance(indices, str):
            indices = [indices]
        to_del = set(indices)
        kept = [bd for bd in self.storage if bd.index not in to_del]
        self.storage.clear()
        self.indices.clear()
        self._pushed = 0
        for bd in kept:
            self.storage.append(bd)
            self.indices.append(bd.index)
            self._pushed += 1

    @apply_middleware("clear")
    def clear(self) -> None:
        self.storage.clear()
        self.indices.clear()
        self._pushed = 0

    def count(self) -> int:
        return len(self.storage)

    def get(self, idx: int) -> BufferedData:
        return self.storage[idx]

    def export_data(self) -> List[BufferedData]:
        return list(self.storage)

    def import_data(self, data: List[BufferedData]) -> None:
        self.clear()
        for bd in data:
            self.storage.append(bd)
            self.indices.append(bd.index)
            self._pushed += 1

    def __iter__(self):
        return iter(self.storage)

    def __len__(self):
        return len(self.storage)
