"""Minimal gym-compatible space classes (the offline image has no gym).

API surface used by the framework: .shape, .dtype, .sample(), .contains(),
plus Discrete.n and Box.low/high.
"""
from typing import Optional, Tuple

import numpy as np


class Space:

    def __init__(self, shape: Optional[Tuple[int, ...]] = None, dtype=None):
        self.shape = tuple(shape) if shape is not None else None
        self.dtype = np.dtype(dtype) if dtype is not None else None
        self._rng = np.random.RandomState()

    def seed(self, seed: int):
        self._rng = np.random.RandomState(seed)

    def sample(self):
        raise NotImplementedError

    def contains(self, x) -> bool:
        raise NotImplementedError


class Discrete(Space):

    def __init__(self, n: int):
        super().__init__(shape=(), dtype=np.int64)
        self.n = int(n)

    def sample(self) -> int:
        return int(self._rng.randint(self.n))

    def contains(self, x) -> bool:
        return 0 <= int(x) < self.n

    def __repr__(self):
        return f"Discrete({self.n})"


class MultiDiscrete(Space):

    def __init__(self, nvec):
        self.nvec = np.asarray(nvec, dtype=np.int64)
        super().__init__(shape=self.nvec.shape, dtype=np.int64)

    def sample(self):
        return (self._rng.random_sample(self.nvec.shape) * self.nvec).astype(np.int64)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return bool(((x >= 0) & (x < self.nvec)).all())


class Box(Space):

    def __init__(self, low, high, shape: Optional[Tuple[int, ...]] = None, dtype=np.float32):
        if shape is None:
            shape = np.broadcast(np.asarray(low), np.asarray(high)).shape
        super().__init__(shape=shape, dtype=dtype)
        self.low = np.broadcast_to(np.asarray(low, dtype=dtype), shape).copy()
        self.high = np.broadcast_to(np.asarray(high, dtype=dtype), shape).copy()

    def sample(self) -> np.ndarray:
        low = np.where(np.isfinite(self.low), self.low, -1.0)
        high = np.where(np.isfinite(self.high), self.high, 1.0)
        return self._rng.uniform(low, high, size=self.shape).astype(self.dtype)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and bool((x >= self.low - 1e-6).all() and (x <= self.high + 1e-6).all())

    def __repr__(self):
        return f"Box{self.shape}"


class Dict(Space):

    def __init__(self, spaces: dict):
        super().__init__()
        self.spaces = spaces

    def sample(self):
        return {k: v.sample() for k, v in self.spaces.items()}

    def contains(self, x) -> bool:
        return all(k in x and s.contains(x[k]) for k, s in self.spaces.items())
