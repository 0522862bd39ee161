"""Rating metrics: Elo and a Gaussian skill rating (TrueSkill-style 1v1
update, self-contained — the trueskill package is unavailable offline).

Parity: reference ding/league/metric.py (EloCalculator:7,
TrueSkillCalculator:109).
"""
import math
from typing import Tuple

import numpy as np


class EloCalculator:

    score = {1: 1.0, 0: 0.5, -1: 0.0}  # win / draw / lose

    @classmethod
    def get_new_rating(cls, rating_a: int, rating_b: int, result: int, k_factor: int = 32,
                       beta: int = 400) -> Tuple[int, int]:
        assert result in (1, 0, -1)
        expect_a = 1.0 / (1 + 10 ** ((rating_b - rating_a) / beta))
        expect_b = 1.0 / (1 + 10 ** ((rating_a - rating_b) / beta))
        new_a = round(rating_a + k_factor * (cls.score[result] - expect_a))
        new_b = round(rating_b + k_factor * (cls.score[-result] - expect_b))
        return new_a, new_b

    @classmethod
    def get_new_rating_array(cls, rating: np.ndarray, result: np.ndarray, match_num: np.ndarray,
                             k_factor: int = 32, beta: int = 400) -> np.ndarray:
        """Batch pairwise update: result[i][j] = cumulative score of i vs j."""
        n = len(rating)
        expect = np.zeros((n, n))
        for i in range(n):
            for j in range(n):
                expect[i][j] = 1.0 / (1 + 10 ** ((rating[j] - rating[i]) / beta)) * match_num[i][j]
        new_rating = rating + k_factor * (result - expect).sum(axis=1)
        return np.round(new_rating).astype(np.int64)


class PlayerRating:
    """Gaussian skill (mu, sigma)."""

    def __init__(self, mu: float = 25.0, sigma: float = 25.0 / 3):
        self.mu = mu
        self.sigma = sigma

    @property
    def exposure(self) -> float:
        return self.mu - 3 * self.sigma

    def __repr__(self):
        return f"Rating(mu={self.mu:.3f}, sigma={self.sigma:.3f})"


class TrueSkillCalculator:
    """Two-player TrueSkill update (Herbrich et al. 2006, no draw margin
    approximation beyond eps)."""

    BETA = 25.0 / 6
    TAU = 25.0 / 300

    @classmethod
    def _v_w(cls, t: float) -> Tuple[float, float]:
        # truncated gaussian moments
        from math import erf, exp, pi, sqrt

        def pdf(x):
            return exp(-x * x / 2) / sqrt(2 * pi)

        def cdf(x):
            return 0.5 * (1 + erf(x / sqrt(2)))

        denom = max(cdf(t), 1e-9)
        v = pdf(t) / denom
        w = v * (v + t)
        return v, w

    @classmethod
    def get_new_rating(cls, a: PlayerRating, b: PlayerRating, result: int) -> Tuple[PlayerRating, PlayerRating]:
        assert result in (1, 0, -1)
        if result == -1:
            b2, a2 = cls.get_new_rating(b, a, 1)
            return a2, b2
        # result 1 (a wins) or 0 (draw ~ treat as tiny win for stability)
        sig_a2 = a.sigma ** 2 + cls.TAU ** 2
        sig_b2 = b.sigma ** 2 + cls.TAU ** 2
        c = math.sqrt(sig_a2 + sig_b2 + 2 * cls.BETA ** 2)
        t = (a.mu - b.mu) / c
        v, w = cls._v_w(t)
        if result == 0:
            v, w = v * 0.5, w * 0.5
        mu_a = a.mu + (sig_a2 / c) * v
        mu_b = b.mu - (sig_b2 / c) * v
        sigma_a = math.sqrt(max(sig_a2 * (1 - (sig_a2 / c ** 2) * w), 1e-6))
        sigma_b = math.sqrt(max(sig_b2 * (1 - (sig_b2 / c ** 2) * w), 1e-6))
        return PlayerRating(mu_a, sigma_a), PlayerRating(mu_b, sigma_b)


class LeagueMetricEnv:
    """Rating environment bound to one metric type."""

    def __init__(self, metric: str = 'trueskill', **kwargs):
        assert metric in ('elo', 'trueskill')
        self.metric = metric

    def create_rating(self, mu: float = 25.0, sigma: float = 25.0 / 3):
        if self.metric == 'elo':
            return 1200
        return PlayerRating(mu, sigma)

    def rate_1vs1(self, a, b, drawn: bool = False):
        if self.metric == 'elo':
            return EloCalculator.get_new_rating(a, b, 0 if drawn else 1)
        return TrueSkillCalculator.get_new_rating(a, b, 0 if drawn else 1)


def get_elo(rating_a, rating_b, result):
    return EloCalculator.get_new_rating(rating_a, rating_b, result)


get_elo_array = EloCalculator.get_new_rating_array
