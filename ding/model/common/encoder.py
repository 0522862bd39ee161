"""Observation encoders.

Parity: reference ding/model/common/encoder.py (ConvEncoder:24, FCEncoder:158,
IMPALAConvEncoder:390, GaussianFourierProjectionTimeEncoder:476).
"""
import math
from typing import Dict, List, Optional, Union

import torch
import torch.nn as nn

from ding.torch_utils import ResBlock, ResFCBlock, conv2d_block, fc_block, build_activation, normed_linear, normed_conv2d


def prod(iterable):
    out = 1
    for x in iterable:
        out *= x
    return out


class UnfoldConv2d(nn.Module):
    """Conv2d computed as one batched unfold + GEMM.

    MI355X note: MIOpen has no winograd/asm solver for large-kernel strided
    convs (e.g. the Atari 8x8 stride-4 first layer) and falls back to a
    PER-SAMPLE im2col+GEMM loop — rocprof on the PPO bench showed 528
    dispatches x batch(320) Im2d2Col launches per step. F.unfold emits ONE
    kernel for the whole batch and the matmul goes to hipBLASLt. Numerically
    identical to nn.Conv2d (same weight layout/state_dict keys).
    """

    def __init__(self, in_channels: int, out_channels: int, kernel_size: int, stride: int, padding: int = 0):
        super().__init__()
        self.conv = nn.Conv2d(in_channels, out_channels, kernel_size, stride, padding)
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape
        k, s, p = self.kernel_size, self.stride, self.padding
        Ho = (H + 2 * p - k) // s + 1
        Wo = (W + 2 * p - k) // s + 1
        cols = torch.nn.functional.unfold(x, k, stride=s, padding=p)  # [B, C*k*k, Ho*Wo]
        w = self.conv.weight.reshape(self.conv.out_channels, -1)  # [O, C*k*k]
        out = torch.matmul(w, cols) + self.conv.bias.reshape(1, -1, 1)  # batched hipBLASLt GEMM
        return out.reshape(B, self.conv.out_channels, Ho, Wo)


class NativeWrwConv2d(nn.Conv2d):
    """nn.Conv2d whose backward-weights runs the hand-written NHWC HIP
    kernel (ding/ops/csrc/wrw_ops.hip) on GPU fp32 channels_last inputs.

    OPT-IN (DING_NATIVE_WRW=1): measured on MI355X the hand-written wrw is
    numerically exact but 2-4x slower than MIOpen's tuned implicit-GEMM
    picks at the Atari shapes (hip 0.16-0.36 ms vs miopen 0.04-0.09 ms,
    tests/test_ops_gpu.py::test_conv_wrw_timing), and the PPO bench
    regressed 17.8k -> 13.4k samples/s with it on. The round-1 naive-fp64
    wrw solver only appears inside MIOpen find-mode contexts and was
    already off the critical path (profiles/README.md), so MIOpen stays
    the default; the kernel remains as the guaranteed-fp32 fallback."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        import os
        if (
            os.environ.get('DING_NATIVE_WRW', '0') in ('1', 'true')
            and x.is_cuda and x.dtype == torch.float32 and torch.is_grad_enabled()
            and self.padding == (0, 0) and self.dilation == (1, 1) and self.groups == 1
            and x.is_contiguous(memory_format=torch.channels_last)
        ):
            from ding.ops import dispatch as _dispatch
            if _dispatch.use_hip_autograd(x):
                return _dispatch.wrw_conv2d(x, self.weight, self.bias, self.stride)
        return super().forward(x)


class StemConv2d(NativeWrwConv2d):
    """Drop-in nn.Conv2d for the Atari stem (8x8, stride 4, pad 0) that
    routes GPU fp32 forwards through the direct HIP kernel
    (ding/ops/csrc/conv_ops.hip) — rocprof showed MIOpen's tuned choice for
    this shape is a per-sample im2col loop (92k Im2d2Col launches per PPO
    bench run). State dict and numerics match nn.Conv2d (fp32 FMA order
    differs within tolerance). Input gradients are not produced: the stem
    is the input layer."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        import os
        if (
            os.environ.get('DING_STEM_CONV', '0') in ('1', 'true')
            and x.is_cuda and x.dtype == torch.float32 and not x.requires_grad
            and self.kernel_size == (8, 8) and self.stride == (4, 4) and self.padding == (0, 0)
            and (x.shape[-1] - 8) // 4 + 1 in (15, 20)
        ):
            from ding.ops import dispatch as _dispatch
            if _dispatch.use_hip_autograd(x):
                return _dispatch.stem_conv2d(x, self.weight, self.bias)
        return super().forward(x)


class ConvEncoder(nn.Module):
    """Nature-DQN style conv stack + flatten + fc to hidden_size_list[-1]."""

    def __init__(
        self,
        obs_shape: tuple,
        hidden_size_list: List[int] = [32, 64, 64, 128],
        activation: str = 'relu',
        kernel_size: List[int] = [8, 4, 3],
        stride: List[int] = [4, 2, 1],
        padding: Optional[List[int]] = None,
        layer_norm: bool = False,
        norm_type: Optional[str] = None,
        fast_im2col: Optional[bool] = None,
    ):
        super().__init__()
        self.obs_shape = obs_shape
        if kernel_size == [8, 4, 3] and stride == [4, 2, 1] and min(obs_shape[1:]) < 36:
            # the Atari stack underflows on small maps (maze/sokoban/procgen
            # crops): fall back to a 3x3 stack that fits >= 5x5 inputs
            kernel_size = [3, 3, 3]
            stride = [2, 2, 1] if min(obs_shape[1:]) >= 12 else [1, 1, 1]
        if padding is None:
            padding = [0] * len(kernel_size)
        if fast_im2col is None:
            import os
            fast_im2col = os.environ.get('DING_FAST_IM2COL', '0') in ('1', 'true')
        layers = []
        in_c = obs_shape[0]
        for i, (k, s, p) in enumerate(zip(kernel_size, stride, padding)):
            if fast_im2col and k >= 5:
                # large-kernel strided conv: batched unfold+GEMM (see UnfoldConv2d)
                layers.append(UnfoldConv2d(in_c, hidden_size_list[i], k, s, p))
                layers.append(build_activation(activation))
            elif i == 0 and k == 8 and s == 4 and p == 0 and norm_type is None:
                # Atari stem: direct HIP conv on GPU (see StemConv2d)
                layers.append(StemConv2d(in_c, hidden_size_list[i], k, s, p))
                layers.append(build_activation(activation))
            else:
                layers.append(
                    conv2d_block(
                        in_c, hidden_size_list[i], k, s, p, activation=activation, norm_type=norm_type,
                        conv_cls=NativeWrwConv2d if p == 0 else None
                    )
                )
            in_c = hidden_size_list[i]
        layers.append(nn.Flatten())
        self.main = nn.Sequential(*layers)
        flatten_size = self._get_flatten_size()
        self.output_size = hidden_size_list[-1]
        self.mid = nn.Linear(flatten_size, hidden_size_list[-1])
        self.act = build_activation(activation)

    def _get_flatten_size(self) -> int:
        with torch.no_grad():
            test = torch.zeros(1, *self.obs_shape)
            return self.main(test).shape[1]

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.act(self.mid(self.main(x)))


class FCEncoder(nn.Module):
    """MLP encoder for vector observations."""

    def __init__(
        self,
        obs_shape: int,
        hidden_size_list: List[int],
        res_block: bool = False,
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        dropout: Optional[float] = None,
    ):
        super().__init__()
        self.obs_shape = obs_shape
        act = activation
        self.init = nn.Linear(obs_shape, hidden_size_list[0])
        self.act = build_activation(activation)
        if res_block:
            assert len(set(hidden_size_list)) == 1, "res_block requires constant width"
            blocks = [ResFCBlock(hidden_size_list[0], activation=act, norm_type=norm_type, dropout=dropout)
                      for _ in range(len(hidden_size_list))]
            self.main = nn.Sequential(*blocks)
        else:
            layers = []
            for i in range(len(hidden_size_list) - 1):
                layers.append(
                    fc_block(hidden_size_list[i], hidden_size_list[i + 1], activation=act, norm_type=norm_type,
                             use_dropout=dropout is not None, dropout_probability=dropout or 0.5)
                )
            self.main = nn.Sequential(*layers)
        self.output_size = hidden_size_list[-1]

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.main(self.act(self.init(x)))


class StructEncoder(nn.Module):
    """Dict-obs encoder: independent sub-encoders concatenated."""

    def __init__(self, encoders: Dict[str, nn.Module]):
        super().__init__()
        self.encoders = nn.ModuleDict(encoders)
        self.output_size = sum(getattr(e, 'output_size', 0) for e in encoders.values())

    def forward(self, x: Dict[str, torch.Tensor]) -> torch.Tensor:
        return torch.cat([self.encoders[k](x[k]) for k in sorted(self.encoders.keys())], dim=-1)


class IMPALACnnResidualBlock(nn.Module):

    def __init__(self, depth: int, scale: float = 1.0, batch_norm: bool = False):
        super().__init__()
        self.scale = scale
        s = math.sqrt(scale)
        self.conv0 = normed_conv2d(depth, depth, 3, padding=1, scale=s)
        self.conv1 = normed_conv2d(depth, depth, 3, padding=1, scale=s)
        self.bn0 = nn.BatchNorm2d(depth) if batch_norm else None
        self.bn1 = nn.BatchNorm2d(depth) if batch_norm else None

    def forward(self, x):
        out = x
        if self.bn0 is not None:
            out = self.bn0(out)
        out = self.conv0(torch.relu(out))
        if self.bn1 is not None:
            out = self.bn1(out)
        out = self.conv1(torch.relu(out))
        return x + out


class IMPALACnnDownStack(nn.Module):
    """conv -> maxpool -> nblock residual blocks."""

    def __init__(self, in_c: int, nblock: int, out_c: int, scale: float = 1.0, pool: bool = True, **kwargs):
        super().__init__()
        self.pool = pool
        self.firstconv = normed_conv2d(in_c, out_c, 3, padding=1)
        s = scale / math.sqrt(nblock)
        self.blocks = nn.ModuleList([IMPALACnnResidualBlock(out_c, scale=s, **kwargs) for _ in range(nblock)])

    def forward(self, x):
        x = self.firstconv(x)
        if self.pool:
            x = torch.nn.functional.max_pool2d(x, kernel_size=3, stride=2, padding=1)
        for b in self.blocks:
            x = b(x)
        return x


class IMPALAConvEncoder(nn.Module):
    """IMPALA resnet encoder (dm-style): 3 down-stacks + fc."""

    name = "IMPALAConvEncoder"

    def __init__(
        self,
        obs_shape: tuple,
        channels: tuple = (16, 32, 32),
        outsize: int = 256,
        scale_ob: float = 255.0,
        nblock: int = 2,
        final_relu: bool = True,
        **kwargs,
    ):
        super().__init__()
        self.scale_ob = scale_ob
        c, h, w = obs_shape
        curshape = (c, h, w)
        s = 1 / math.sqrt(len(channels))
        self.stacks = nn.ModuleList()
        for out_c in channels:
            stack = IMPALACnnDownStack(curshape[0], nblock=nblock, out_c=out_c, scale=s, **kwargs)
            self.stacks.append(stack)
            curshape = (out_c, (curshape[1] + 1) // 2, (curshape[2] + 1) // 2)
        self.dense = normed_linear(prod(curshape), outsize, scale=1.4)
        self.outsize = outsize
        self.output_size = outsize
        self.final_relu = final_relu

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.float() / self.scale_ob
        for stack in self.stacks:
            x = stack(x)
        x = x.reshape(x.shape[0], -1)
        x = torch.relu(x)
        x = self.dense(x)
        if self.final_relu:
            x = torch.relu(x)
        return x


class GaussianFourierProjectionTimeEncoder(nn.Module):
    """Random-Fourier time embedding (diffusion models)."""

    def __init__(self, embed_dim: int, scale: float = 30.0):
        super().__init__()
        self.W = nn.Parameter(torch.randn(embed_dim // 2) * scale, requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x_proj = x[..., None] * self.W[None, :] * 2 * math.pi
        return torch.cat([torch.sin(x_proj), torch.cos(x_proj)], dim=-1)
