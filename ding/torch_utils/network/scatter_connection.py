"""Scatter entity features onto a spatial map (AlphaStar spatial encoder).

Parity: reference ding/torch_utils/network/scatter_connection.py:30
(ScatterConnection.forward:59, xy_forward:90). DI-hpc kernel #7 in SURVEY
§2.9a — on GPU this dispatches to the HIP scatter kernel (atomic add /
ordered cover semantics).
"""
from typing import Tuple

import torch
import torch.nn as nn


class ScatterConnection(nn.Module):

    def __init__(self, scatter_type: str):
        super().__init__()
        assert scatter_type in ('cover', 'add')
        self.scatter_type = scatter_type

    def forward(self, x: torch.Tensor, spatial_size: Tuple[int, int], location: torch.Tensor) -> torch.Tensor:
        """x [B,M,N] entity features; location [B,M,2] (y, x) coords ->
        output [B,N,H,W]."""
        from ding.ops import dispatch
        device, dtype = x.device, x.dtype
        B, M, N = x.shape
        H, W = spatial_size
        index = location[..., 0] * W + location[..., 1]  # [B, M]
        if dispatch.use_hip(x):
            return dispatch.scatter_connection(x, index, H, W, self.scatter_type)
        index = index.long().unsqueeze(-1).expand(B, M, N)  # [B, M, N]
        output = torch.zeros(B, H * W, N, device=device, dtype=dtype)
        if self.scatter_type == 'cover':
            output.scatter_(dim=1, index=index, src=x)
        else:
            output.scatter_add_(dim=1, index=index, src=x)
        return output.permute(0, 2, 1).reshape(B, N, H, W)

    def xy_forward(
        self, x: torch.Tensor, spatial_size: Tuple[int, int], coord_x: torch.Tensor, coord_y: torch.Tensor
    ) -> torch.Tensor:
        location = torch.stack([coord_x, coord_y], dim=-1)
        return self.forward(x, spatial_size, location)
