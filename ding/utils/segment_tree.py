"""Sum/Min segment trees backing prioritized experience replay.

Parity: reference ding/utils/segment_tree.py (SegmentTree:37,
SumSegmentTree:133, MinSegmentTree:168) whose hot kernels are numba-JIT
(:187,210,247). Here the hot path is the in-tree C++ extension
``ding.utils._ctree`` (see csrc/ctree.cpp); a pure-numpy fallback keeps the
module importable before ``setup.py build_ext``.
"""
from typing import Optional

import numpy as np

try:
    from ding.utils import _ctree  # C++ extension (built in-tree)
except ImportError:  # pragma: no cover - fallback lane
    _ctree = None

_OP = {"sum": 0, "min": 1, "max": 2}
_NEUTRAL = {"sum": 0.0, "min": float("inf"), "max": -float("inf")}
_NPFN = {"sum": np.add, "min": np.minimum, "max": np.maximum}


class SegmentTree:

    def __init__(self, capacity: int, operation: str, neutral_element: Optional[float] = None):
        assert capacity > 0 and capacity & (capacity - 1) == 0, "capacity must be a power of 2"
        assert operation in _OP
        self.capacity = capacity
        self.operation = operation
        self.neutral_element = _NEUTRAL[operation] if neutral_element is None else neutral_element
        self.value = np.full(2 * capacity, self.neutral_element, dtype=np.float64)
        self._op_i = _OP[operation]
        self._npfn = _NPFN[operation]

    def reduce(self, start: int = 0, end: Optional[int] = None) -> float:
        if end is None:
            end = self.capacity
        if end <= 0:
            end += self.capacity
        assert 0 <= start < end <= self.capacity
        if _ctree is not None:
            return _ctree.reduce_range(self.value, start, end, self._op_i, self.neutral_element)
        # numpy fallback: iterative two-pointer reduce
        result = self.neutral_element
        l, r = start + self.capacity, end + self.capacity
        while l < r:
            if l & 1:
                result = self._npfn(result, self.value[l])
                l += 1
            if r & 1:
                r -= 1
                result = self._npfn(result, self.value[r])
            l >>= 1
            r >>= 1
        return float(result)

    def __setitem__(self, idx, val) -> None:
        idx = np.atleast_1d(np.asarray(idx, dtype=np.int64))
        val = np.atleast_1d(np.asarray(val, dtype=np.float64))
        assert idx.shape == val.shape
        assert ((0 <= idx) & (idx < self.capacity)).all()
        if _ctree is not None:
            _ctree.setitem_batch(self.value, idx, val, self._op_i)
            return
        for i, v in zip(idx, val):
            node = int(i) + self.capacity
            self.value[node] = v
            node //= 2
            while node >= 1:
                self.value[node] = self._npfn(self.value[2 * node], self.value[2 * node + 1])
                node //= 2

    def __getitem__(self, idx):
        if isinstance(idx, (int, np.integer)):
            assert 0 <= idx < self.capacity
            return float(self.value[int(idx) + self.capacity])
        idx = np.asarray(idx, dtype=np.int64)
        return self.value[idx + self.capacity].copy()


class SumSegmentTree(SegmentTree):

    def __init__(self, capacity: int):
        super().__init__(capacity, "sum")

    def find_prefixsum_idx(self, prefixsum, trust_caller: bool = True):
        """Leaf index i such that sum(value[:i]) <= prefixsum < sum(value[:i+1]).

        Accepts a scalar or a batch array (batched path is the PER sampling
        hot loop; reference advanced_buffer.py:536).
        """
        scalar = np.isscalar(prefixsum)
        p = np.atleast_1d(np.asarray(prefixsum, dtype=np.float64))
        if not trust_caller:
            total = self.reduce()
            assert ((0 <= p) & (p < total + 1e-5)).all()
        if _ctree is not None:
            out = _ctree.find_prefixsum_idx_batch(self.value, p)
        else:
            out = np.empty(p.shape[0], dtype=np.int64)
            for j, rem in enumerate(p):
                node = 1
                while node < self.capacity:
                    left = 2 * node
                    if self.value[left] > rem:
                        node = left
                    else:
                        rem -= self.value[left]
                        node = left + 1
                out[j] = node - self.capacity
        return int(out[0]) if scalar else out


class MinSegmentTree(SegmentTree):

    def __init__(self, capacity: int):
        super().__init__(capacity, "min")
