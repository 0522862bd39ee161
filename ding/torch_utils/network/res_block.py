"""Residual blocks (conv + fc variants).

Parity: reference ding/torch_utils/network/res_block.py (ResBlock, ResFCBlock)
used by IMPALA encoder and AlphaStar-style nets.
"""
from typing import Optional

import torch
import torch.nn as nn

from .nn_module import conv2d_block, fc_block, build_activation


class ResBlock(nn.Module):
    """2-conv residual block; res_type in {'basic','bottleneck','downsample'}."""

    def __init__(
        self,
        in_channels: int,
        activation: str = 'relu',
        norm_type: Optional[str] = 'BN',
        res_type: str = 'basic',
        bias: bool = True,
        out_channels: Optional[int] = None,
    ):
        super().__init__()
        self.act = build_activation(activation)
        assert res_type in ('basic', 'bottleneck', 'downsample')
        self.res_type = res_type
        out_channels = out_channels or in_channels
        if res_type == 'basic':
            self.conv1 = conv2d_block(in_channels, out_channels, 3, 1, 1, activation=activation, norm_type=norm_type, bias=bias)
            self.conv2 = conv2d_block(out_channels, out_channels, 3, 1, 1, activation=None, norm_type=norm_type, bias=bias)
        elif res_type == 'bottleneck':
            mid = out_channels // 4
            self.conv1 = conv2d_block(in_channels, mid, 1, 1, 0, activation=activation, norm_type=norm_type, bias=bias)
            self.conv2 = conv2d_block(mid, mid, 3, 1, 1, activation=activation, norm_type=norm_type, bias=bias)
            self.conv3 = conv2d_block(mid, out_channels, 1, 1, 0, activation=None, norm_type=norm_type, bias=bias)
        else:  # downsample
            self.conv1 = conv2d_block(in_channels, out_channels, 3, 2, 1, activation=activation, norm_type=norm_type, bias=bias)
            self.conv2 = conv2d_block(out_channels, out_channels, 3, 1, 1, activation=None, norm_type=norm_type, bias=bias)
            self.shortcut = conv2d_block(in_channels, out_channels, 3, 2, 1, activation=None, norm_type=None, bias=bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        if self.res_type == 'basic':
            out = self.conv2(self.conv1(x))
        elif self.res_type == 'bottleneck':
            out = self.conv3(self.conv2(self.conv1(x)))
        else:
            out = self.conv2(self.conv1(x))
            identity = self.shortcut(x)
        return self.act(out + identity)


class ResFCBlock(nn.Module):
    """2-fc residual block."""

    def __init__(self, in_channels: int, activation: str = 'relu', norm_type: Optional[str] = 'LN', dropout: Optional[float] = None):
        super().__init__()
        self.act = build_activation(activation)
        self.fc1 = fc_block(in_channels, in_channels, activation=activation, norm_type=norm_type)
        self.fc2 = fc_block(in_channels, in_channels, activation=None, norm_type=norm_type)
        self.dropout = nn.Dropout(dropout) if dropout is not None else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.fc2(self.fc1(x))
        out = self.act(x + out)
        if self.dropout is not None:
            out = self.dropout(out)
        return out
