"""GPU-resident prioritized replay buffer.

MI355X design: transitions live as pre-collated tensors in the 288 GB HBM3E
(no per-sample python objects, no host round-trips); priorities are a device
vector and prioritized sampling is inverse-CDF via ``torch.cumsum`` +
``torch.searchsorted`` — two library kernels instead of a host-side
segment tree walked per sample (SURVEY §7 "PER sum-tree on GPU").

The ring is allocated lazily from the first pushed batch's schema; pushes,
samples and priority updates are all batched tensor ops on the buffer's
device, so a learner can sample minibatches without leaving the GPU.
"""
from typing import Dict, Optional, Tuple

import torch


class GPUPrioritizedBuffer:
    """Flat-tensor PER ring. API: push(dict_of_tensors) / sample(n) ->
    (batch, indices, is_weights) / update_priority(indices, priorities)."""

    def __init__(
        self,
        size: int,
        alpha: float = 0.6,
        beta: float = 0.4,
        eps: float = 0.01,
        device: str = 'cuda',
    ):
        self.size = size
        self.alpha = alpha
        self.beta = beta
        self.eps = eps
        self.device = device
        self._storage: Optional[Dict[str, torch.Tensor]] = None
        self._priority = torch.zeros(size, device=device)
        self._max_priority = 1.0
        self._tail = 0
        self._count = 0

    def _allocate(self, batch: Dict[str, torch.Tensor]) -> None:
        self._storage = {
            k: torch.empty(self.size, *v.shape[1:], dtype=v.dtype, device=self.device)
            for k, v in batch.items()
        }

    def push(self, batch: Dict[str, torch.Tensor], priorities: Optional[torch.Tensor] = None) -> torch.Tensor:
        """batch: {key: [B, ...]} on any device; returns the ring indices."""
        B = next(iter(batch.values())).shape[0]
        if self._storage is None:
            self._allocate(batch)
        idx = (torch.arange(B, device=self.device) + self._tail) % self.size
        for k, v in batch.items():
            self._storage[k][idx] = v.to(self.device, non_blocking=True)
        if priorities is None:
            self._priority[idx] = self._max_priority
        else:
            p = priorities.to(self.device).clamp_min(self.eps)
            self._priority[idx] = p ** self.alpha
        self._tail = int((self._tail + B) % self.size)
        self._count = min(self._count + B, self.size)
        return idx

    def sample(self, n: int, beta: Optional[float] = None) -> Tuple[Dict[str, torch.Tensor], torch.Tensor,
                                                                    torch.Tensor]:
        """Inverse-CDF prioritized sampling, fully on device."""
        assert self._count > 0, "empty buffer"
        beta = self.beta if beta is None else beta
        p = self._priority[:self._count]
        cdf = torch.cumsum(p, dim=0)
        total = cdf[-1]
        u = torch.rand(n, device=self.device) * total
        idx = torch.searchsorted(cdf, u).clamp_max_(self._count - 1)
        probs = p[idx] / total
        weights = (self._count * probs).pow(-beta)
        weights = weights / weights.max()
        batch = {k: v[idx] for k, v in self._storage.items()}
        return batch, idx, weights

    def update_priority(self, indices: torch.Tensor, priorities: torch.Tensor) -> None:
        p = priorities.detach().to(self.device).clamp_min(self.eps)
        self._priority[indices] = p ** self.alpha
        self._max_priority = max(self._max_priority, float(p.max()))

    def count(self) -> int:
        return self._count

    def clear(self) -> None:
        self._priority.zero_()
        self._tail = 0
        self._count = 0

    def state_dict(self) -> dict:
        return {
            'storage': self._storage, 'priority': self._priority, 'tail': self._tail,
            'count': self._count, 'max_priority': self._max_priority,
        }

    def load_state_dict(self, sd: dict) -> None:
        self._storage = sd['storage']
        self._priority = sd['priority']
        self._tail = sd['tail']
        self._count = sd['count']
        self._max_priority = sd['max_priority']
