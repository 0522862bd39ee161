"""Math helpers. Parity: reference ding/torch_utils/math_helper.py (cov)."""
from typing import Optional

import torch


def cov(
    x: torch.Tensor,
    rowvar: bool = False,
    bias: bool = False,
    ddof: Optional[int] = None,
    aweights: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """numpy.cov-compatible covariance matrix for 2D tensors."""
    x = x if rowvar else x.t()
    if x.dim() == 1:
        x = x.unsqueeze(0)
    if ddof is None:
        ddof = 0 if bias else 1
    w = aweights
    if w is not None:
        w = w / w.sum()
        avg = (x * w.unsqueeze(0)).sum(dim=1)
    else:
        avg = x.mean(dim=1)
    if w is None:
        fact = x.shape[1] - ddof
        xm = x - avg.unsqueeze(1)
        return (xm @ xm.t()) / fact
    w_sum = torch.ones((), device=x.device)  # already normalized
    if ddof == 0:
        fact = w_sum
    elif aweights is None:
        fact = w_sum - ddof
    else:
        fact = w_sum - ddof * (w * w).sum() / w_sum
    xm = x - avg.unsqueeze(1)
    return (xm * w.unsqueeze(0)) @ xm.t() / fact


def unsqueeze_repeat(x: torch.Tensor, repeat_times: int, unsqueeze_dim: int = 0) -> torch.Tensor:
    """Insert a dim and repeat along it (MBPO ensemble broadcasting)."""
    repeats = [1] * (x.dim() + 1)
    repeats[unsqueeze_dim] = repeat_times
    return x.unsqueeze(unsqueeze_dim).repeat(*repeats)
