"""Env-side feature encoders + action transforms.

Parity: reference ding/envs/common/common_function.py (sqrt_one_hot:14,
div_one_hot:31, clip_one_hot:66, batch_binary_encode:184,
affine_transform:243, get_postion_vector:225).
"""
from typing import Any, Optional

import numpy as np
import torch


def one_hot(v: torch.Tensor, num: int) -> torch.Tensor:
    return torch.nn.functional.one_hot(v.long().clamp(0, num - 1), num).float()


def sqrt_one_hot(v: torch.Tensor, max_val: int) -> torch.Tensor:
    """One-hot of floor(sqrt(v)) — compresses large count features."""
    num = int(np.floor(np.sqrt(max_val))) + 1
    v = torch.sqrt(torch.clamp(v.float(), 0, max_val)).floor().long()
    return one_hot(v, num)


def div_one_hot(v: torch.Tensor, max_val: int, ratio: int) -> torch.Tensor:
    """One-hot of v // ratio."""
    num = max_val // ratio + 1
    v = torch.clamp(v.float(), 0, max_val).div(ratio).floor().long()
    return one_hot(v, num)


def div_func(inputs: torch.Tensor, other: float, unsqueeze_dim: int = 1) -> torch.Tensor:
    out = inputs.float() / other
    return out.unsqueeze(unsqueeze_dim) if unsqueeze_dim is not None else out


def clip_one_hot(v: torch.Tensor, num: int) -> torch.Tensor:
    return one_hot(torch.clamp(v.long(), 0, num - 1), num)


def batch_binary_encode(x: torch.Tensor, bit_num: int) -> torch.Tensor:
    """Integer -> fixed-width binary feature rows."""
    x = x.long()
    masks = 2 ** torch.arange(bit_num - 1, -1, -1, device=x.device, dtype=torch.long)
    return x.unsqueeze(-1).bitwise_and(masks).ne(0).float()


def get_postion_vector(x: list) -> torch.Tensor:
    """Transformer-style sinusoidal position encoding of a 32-bit key."""
    v = torch.zeros(64, dtype=torch.float)
    x = torch.as_tensor(x, dtype=torch.float)
    div = torch.exp(torch.arange(0, 64, 2, dtype=torch.float) * (-np.log(10000.0) / 64))
    v[0::2] = torch.sin(x.sum() * div)
    v[1::2] = torch.cos(x.sum() * div)
    return v


def affine_transform(
    data: Any,
    action_clip: bool = True,
    alpha: Optional[float] = None,
    beta: Optional[float] = None,
    min_val: Optional[float] = None,
    max_val: Optional[float] = None,
) -> Any:
    """Map [-1, 1] actions to [min_val, max_val] (or alpha*x + beta)."""
    if action_clip:
        data = np.clip(data, -1, 1)
    if min_val is not None:
        assert max_val is not None
        alpha = (max_val - min_val) / 2
        beta = (max_val + min_val) / 2
    assert alpha is not None
    beta = beta if beta is not None else 0.
    return data * alpha + beta
