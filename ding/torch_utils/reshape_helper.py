"""Batch-dim folding for time-batched tensors.

Parity: reference ding/torch_utils/reshape_helper.py (fold_batch,
unfold_batch, unsqueeze_repeat).
"""
from typing import Tuple, Union

import torch


def fold_batch(x: torch.Tensor, nonbatch_ndims: int = 1) -> Tuple[torch.Tensor, Tuple]:
    """[T, B, ...] -> [T*B, ...]; returns (folded, batch_dims)."""
    if nonbatch_ndims > 0:
        batch_dims = x.shape[:-nonbatch_ndims]
        return x.reshape(-1, *x.shape[-nonbatch_ndims:]), batch_dims
    batch_dims = x.shape
    return x.reshape(-1), batch_dims


def unfold_batch(x: torch.Tensor, batch_dims: Union[Tuple, torch.Size]) -> torch.Tensor:
    return x.reshape(*batch_dims, *x.shape[1:])
