"""Dataset normalizers (Gaussian / Limits / CDF) for offline-RL pipelines.

Parity: reference ding/utils/normalizer_helper.py:4 (DatasetNormalizer and the
per-key normalizer classes used by diffuser/DT datasets).
"""
import numpy as np


class _BaseNorm:

    def __init__(self, x: np.ndarray):
        self.x = x.astype(np.float32).reshape(-1, x.shape[-1])
        self.mins = self.x.min(0)
        self.maxs = self.x.max(0)

    def normalize(self, x):
        raise NotImplementedError

    def unnormalize(self, x):
        raise NotImplementedError


class GaussianNormalizer(_BaseNorm):

    def __init__(self, x):
        super().__init__(x)
        self.means = self.x.mean(0)
        self.stds = self.x.std(0)
        self.stds[self.stds == 0] = 1.0

    def normalize(self, x):
        return (x - self.means) / self.stds

    def unnormalize(self, x):
        return x * self.stds + self.means


class LimitsNormalizer(_BaseNorm):
    """Map to [-1, 1]."""

    def normalize(self, x):
        span = self.maxs - self.mins
        span[span == 0] = 1.0
        return 2.0 * (x - self.mins) / span - 1.0

    def unnormalize(self, x, eps: float = 1e-4):
        x = np.clip(x, -1.0 - eps, 1.0 + eps)
        return (x + 1.0) / 2.0 * (self.maxs - self.mins) + self.mins


class CDFNormalizer(_BaseNorm):
    """Per-dimension empirical-CDF normalization to [0, 1]."""

    def __init__(self, x):
        super().__init__(x)
        self._sorted = np.sort(self.x, axis=0)
        self._n = self._sorted.shape[0]

    def normalize(self, x):
        out = np.empty_like(x, dtype=np.float32)
        flat = x.reshape(-1, x.shape[-1])
        res = np.empty_like(flat, dtype=np.float32)
        for d in range(flat.shape[-1]):
            res[:, d] = np.searchsorted(self._sorted[:, d], flat[:, d]) / self._n
        return res.reshape(x.shape)

    def unnormalize(self, x):
        flat = np.clip(x.reshape(-1, x.shape[-1]), 0, 1)
        res = np.empty_like(flat, dtype=np.float32)
        idx = (flat * (self._n - 1)).astype(np.int64)
        for d in range(flat.shape[-1]):
            res[:, d] = self._sorted[idx[:, d], d]
        return res.reshape(x.shape)


_NORMS = {"gaussian": GaussianNormalizer, "limits": LimitsNormalizer, "cdf": CDFNormalizer}


class DatasetNormalizer:

    def __init__(self, dataset: dict, normalizer: str = "limits", path_lengths=None):
        cls = _NORMS[normalizer] if isinstance(normalizer, str) else normalizer
        self.normalizers = {k: cls(v) for k, v in dataset.items() if isinstance(v, np.ndarray) and v.ndim >= 2}

    def normalize(self, x, key: str):
        return self.normalizers[key].normalize(x)

    def unnormalize(self, x, key: str):
        return self.normalizers[key].unnormalize(x)
