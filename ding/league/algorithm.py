"""Opponent-sampling distributions.

Parity: reference ding/league/algorithm.py:4 (pfsp), uniform.
"""
import numpy as np


def pfsp(win_rates: np.ndarray, weighting: str = 'squared') -> np.ndarray:
    """Prioritized fictitious self-play weights over opponents given my win
    rates against them."""
    weightings = {
        'variance': lambda x: x * (1 - x),
        'linear': lambda x: 1 - x,
        'linear_capped': lambda x: np.minimum(0.5, 1 - x),
        'squared': lambda x: (1 - x) ** 2,
    }
    fn = weightings[weighting]
    probs = fn(np.asarray(win_rates, dtype=np.float64))
    norm = probs.sum()
    if norm < 1e-10:
        return np.full_like(probs, 1 / len(probs))
    return probs / norm


def uniform(win_rates: np.ndarray) -> np.ndarray:
    return np.full(len(win_rates), 1 / len(win_rates))
