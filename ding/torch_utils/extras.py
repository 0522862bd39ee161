"""Small torch utilities completing the reference surface.

Parity: reference ding/torch_utils/backend_helper.py (enable_tf32),
metric.py (levenshtein/hamming), model_helper.py (get_num_params),
nn_test_helper.py (is_differentiable), parameter.py (NonegativeParameter,
TanhParameter), distribution.py (CategoricalPd family),
dataparallel.py (DataParallel).
"""
from typing import Optional, Union

import torch
import torch.nn as nn


def enable_tf32() -> None:
    """Allow reduced-precision matmul accumulation. On ROCm/MI355X this maps
    to hipBLASLt's xf32 paths where available."""
    torch.backends.cudnn.allow_tf32 = True
    torch.backends.cuda.matmul.allow_tf32 = True


def get_num_params(model: nn.Module) -> int:
    return sum(p.numel() for p in model.parameters())


def levenshtein_distance(
    pred: torch.LongTensor, target: torch.LongTensor, pred_extra=None, target_extra=None, extra_fn=None
) -> torch.FloatTensor:
    """Edit distance between two 1-D int sequences (optionally weighting
    substitutions via extra_fn on aligned extras)."""
    p, t = pred.tolist(), target.tolist()
    m, n = len(p), len(t)
    dp = [[0.0] * (n + 1) for _ in range(m + 1)]
    for i in range(m + 1):
        dp[i][0] = i
    for j in range(n + 1):
        dp[0][j] = j
    for i in range(1, m + 1):
        for j in range(1, n + 1):
            if p[i - 1] == t[j - 1]:
                cost = 0.0
                if extra_fn is not None and pred_extra is not None and target_extra is not None:
                    cost = float(extra_fn(pred_extra[i - 1], target_extra[j - 1]))
                dp[i][j] = dp[i - 1][j - 1] + cost
            else:
                dp[i][j] = 1 + min(dp[i - 1][j], dp[i][j - 1], dp[i - 1][j - 1])
    return torch.as_tensor(dp[m][n], dtype=torch.float32)


def hamming_distance(pred: torch.LongTensor, target: torch.LongTensor, weight=1.) -> torch.LongTensor:
    """Row-wise weighted hamming distance of two [B, N] int tensors."""
    assert pred.shape == target.shape
    return ((pred != target).float() * weight).sum(dim=-1).long()


def is_differentiable(loss: torch.Tensor, model: Union[nn.Module, list], print_instead: bool = False) -> None:
    """Assert every parameter receives a gradient from loss.backward()."""
    models = model if isinstance(model, (list, tuple)) else [model]
    for m in models:
        for p in m.parameters():
            assert p.grad is None
    loss.backward()
    for m in models:
        for name, p in m.named_parameters():
            if p.grad is None:
                msg = f'parameter {name} got no gradient'
                if print_instead:
                    print(msg)
                else:
                    raise AssertionError(msg)


class NonegativeParameter(nn.Module):
    """Parameter constrained positive via exp(log_value)."""

    def __init__(self, data: Optional[torch.Tensor] = None, requires_grad: bool = True, delta: float = 1e-8):
        super().__init__()
        if data is None:
            data = torch.ones(1)
        self.log_data = nn.Parameter(torch.log(data + delta), requires_grad=requires_grad)

    @property
    def data(self) -> torch.Tensor:
        return torch.exp(self.log_data)

    def forward(self) -> torch.Tensor:
        return torch.exp(self.log_data)

    def set_data(self, data: torch.Tensor) -> None:
        self.log_data = nn.Parameter(torch.log(data + 1e-8), requires_grad=self.log_data.requires_grad)


class TanhParameter(nn.Module):
    """Parameter constrained to (-1, 1) via tanh(raw)."""

    def __init__(self, data: Optional[torch.Tensor] = None, requires_grad: bool = True):
        super().__init__()
        if data is None:
            data = torch.zeros(1)
        self.data_inv = nn.Parameter(torch.atanh(data.clamp(-0.999999, 0.999999)), requires_grad=requires_grad)

    @property
    def data(self) -> torch.Tensor:
        return torch.tanh(self.data_inv)

    def forward(self) -> torch.Tensor:
        return torch.tanh(self.data_inv)

    def set_data(self, data: torch.Tensor) -> None:
        self.data_inv = nn.Parameter(
            torch.atanh(data.clamp(-0.999999, 0.999999)), requires_grad=self.data_inv.requires_grad
        )


class CategoricalPd:
    """Categorical policy distribution helper (logits in, neglogp/entropy/
    sample out) — reference distribution.py CategoricalPd."""

    def __init__(self, logits: Optional[torch.Tensor] = None):
        self.logits = logits

    def update_logits(self, logits: torch.Tensor) -> None:
        self.logits = logits

    def neglogp(self, x: torch.Tensor, reduction: str = 'mean') -> torch.Tensor:
        nll = torch.nn.functional.cross_entropy(self.logits, x.long(), reduction=reduction)
        return nll

    def entropy(self, reduction: str = 'mean') -> torch.Tensor:
        logp = torch.log_softmax(self.logits, dim=-1)
        ent = -(logp.exp() * logp).sum(-1)
        if reduction == 'mean':
            return ent.mean()
        return ent

    def sample(self) -> torch.Tensor:
        # gumbel-max: sync-free sampling
        g = -torch.log(-torch.log(torch.rand_like(self.logits) + 1e-10) + 1e-10)
        return (self.logits + g).argmax(dim=-1)

    def mode(self) -> torch.Tensor:
        return self.logits.argmax(dim=-1)


class CategoricalPdPytorch(torch.distributions.Categorical):
    """torch Categorical with the Pd-style update interface."""

    def __init__(self, probs=None):
        if probs is not None:
            super().__init__(probs=probs)

    def update_logits(self, logits: torch.Tensor) -> None:
        super().__init__(logits=logits)

    def update_probs(self, probs: torch.Tensor) -> None:
        super().__init__(probs=probs)

    def sample_dot(self) -> torch.Tensor:
        return super().sample()


class DataParallel(nn.DataParallel):
    """nn.DataParallel that forwards attribute access to the wrapped module
    (reference dataparallel.py). NOTE: on MI355X prefer one process per GPU
    over RCCL (ding/parallel) — this exists for API compatibility."""

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self.module, name)
