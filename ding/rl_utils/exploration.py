"""Exploration schedules and action noise.

Parity: reference ding/rl_utils/exploration.py (get_epsilon_greedy_fn,
GaussianNoise, OUNoise, create_noise_generator).
"""
import math
from abc import ABC, abstractmethod
from typing import Callable, Optional

import torch


def get_epsilon_greedy_fn(start: float, end: float, decay: int, type_: str = 'exp') -> Callable:
    """Return eps(step): exp decay end + (start-end)*exp(-step/decay), or
    linear interpolation hitting ``end`` at ``decay`` steps."""
    assert type_ in ('linear', 'exp'), type_
    if type_ == 'exp':
        return lambda x: (start - end) * math.exp(-1.0 * x / decay) + end

    def eps_fn(x):
        if x >= decay:
            return end
        return start - (start - end) * x / decay

    return eps_fn


class BaseNoise(ABC):

    @abstractmethod
    def __call__(self, shape: tuple, device: str) -> torch.Tensor:
        raise NotImplementedError


class GaussianNoise(BaseNoise):

    def __init__(self, mu: float = 0.0, sigma: float = 1.0):
        assert sigma >= 0
        self._mu = mu
        self._sigma = sigma

    def __call__(self, shape: tuple, device: str) -> torch.Tensor:
        return torch.randn(shape, device=device) * self._sigma + self._mu


class OUNoise(BaseNoise):
    """Ornstein-Uhlenbeck process noise with per-call state."""

    def __init__(self, mu: float = 0.0, sigma: float = 0.3, theta: float = 0.15, dt: float = 1e-2, x0=None):
        self._mu = mu
        self._sigma = sigma
        self._theta = theta
        self._dt = dt
        self.x0 = x0
        self._x = None

    def reset(self) -> None:
        self._x = self.x0

    def __call__(self, shape: tuple, device: str, mu: Optional[float] = None) -> torch.Tensor:
        if self._x is None or (isinstance(self._x, torch.Tensor) and self._x.shape != shape):
            self._x = torch.zeros(shape, device=device)
        if mu is None:
            mu = self._mu
        noise = self._x + self._theta * (mu - self._x) * self._dt + \
            self._sigma * math.sqrt(self._dt) * torch.randn(shape, device=device)
        self._x = noise
        return noise


noise_mapping = {'gauss': GaussianNoise, 'ou': OUNoise}


def create_noise_generator(noise_type: str, noise_kwargs: dict) -> BaseNoise:
    if noise_type not in noise_mapping:
        raise KeyError(f"unknown noise type: {noise_type}")
    return noise_mapping[noise_type](**(noise_kwargs or {}))
