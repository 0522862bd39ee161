"""Basic NN building blocks: MLP, conv blocks, normalization/activation
factories, weight init, noisy linear.

Parity: reference ding/torch_utils/network/nn_module.py (MLP, fc_block,
conv2d_block, one_hot, NoisyLinearLayer, ...).
"""
import math
from typing import Callable, List, Optional, Union

import torch
import torch.nn as nn
import torch.nn.functional as F


def weight_init_(weight: torch.Tensor, init_type: Optional[str] = None, activation: Optional[str] = None) -> None:
    if init_type is None:
        return
    if init_type == "xavier":
        nn.init.xavier_uniform_(weight)
    elif init_type == "kaiming":
        nn.init.kaiming_uniform_(weight, nonlinearity=activation or "relu")
    elif init_type == "orthogonal":
        nn.init.orthogonal_(weight)
    elif init_type == "zero":
        nn.init.zeros_(weight)
    else:
        raise KeyError(f"unknown init type: {init_type}")


def build_activation(activation: Union[str, nn.Module, None], inplace: bool = False) -> Optional[nn.Module]:
    if activation is None or isinstance(activation, nn.Module):
        return activation
    table = {
        "relu": lambda: nn.ReLU(inplace=inplace),
        "leaky_relu": lambda: nn.LeakyReLU(0.01, inplace=inplace),
        "gelu": nn.GELU,
        "tanh": nn.Tanh,
        "sigmoid": nn.Sigmoid,
        "softplus": nn.Softplus,
        "elu": nn.ELU,
        "square": lambda: Lambda(lambda x: x ** 2),
        "identity": nn.Identity,
        "silu": nn.SiLU,
    }
    if activation not in table:
        raise KeyError(f"unknown activation: {activation}")
    return table[activation]()


def build_normalization(norm_type: Optional[str], dim: Optional[int] = None) -> Optional[Callable]:
    """Return a normalization layer constructor. dim selects 1d/2d variants."""
    if norm_type is None:
        return None
    if norm_type == "BN":
        return nn.BatchNorm2d if dim == 2 else nn.BatchNorm1d
    if norm_type == "LN":
        return nn.LayerNorm
    if norm_type == "IN":
        return nn.InstanceNorm2d if dim == 2 else nn.InstanceNorm1d
    if norm_type == "GN":
        return nn.GroupNorm
    if norm_type == "SyncBN":
        return nn.SyncBatchNorm
    raise KeyError(f"unknown norm type: {norm_type}")


class Lambda(nn.Module):

    def __init__(self, fn: Callable):
        super().__init__()
        self.fn = fn

    def forward(self, x):
        return self.fn(x)


def sequential_pack(layers: List[nn.Module]) -> nn.Sequential:
    return nn.Sequential(*[l for l in layers if l is not None])


def fc_block(
    in_channels: int,
    out_channels: int,
    activation: Union[str, nn.Module, None] = None,
    norm_type: Optional[str] = None,
    use_dropout: bool = False,
    dropout_probability: float = 0.5,
    bias: bool = True,
) -> nn.Sequential:
    layers = [nn.Linear(in_channels, out_channels, bias=bias)]
    if norm_type is not None:
        layers.append(build_normalization(norm_type, dim=1)(out_channels))
    layers.append(build_activation(activation, inplace=False))
    if use_dropout:
        layers.append(nn.Dropout(dropout_probability))
    return sequential_pack(layers)


def conv2d_block(
    in_channels: int,
    out_channels: int,
    kernel_size: int,
    stride: int = 1,
    padding: int = 0,
    dilation: int = 1,
    groups: int = 1,
    pad_type: str = "zero",
    activation: Union[str, nn.Module, None] = None,
    norm_type: Optional[str] = None,
    num_groups_for_gn: int = 1,
    bias: bool = True,
    conv_cls: type = None,
) -> nn.Sequential:
    conv_cls = conv_cls or nn.Conv2d
    layers = []
    if pad_type == "zero":
        layers.append(
            conv_cls(in_channels, out_channels, kernel_size, stride, padding, dilation, groups, bias=bias)
        )
    elif pad_type in ("reflect", "replicate"):
        pad_cls = nn.ReflectionPad2d if pad_type == "reflect" else nn.ReplicationPad2d
        layers.append(pad_cls(padding))
        layers.append(conv_cls(in_channels, out_channels, kernel_size, stride, 0, dilation, groups, bias=bias))
    else:
        raise KeyError(pad_type)
    if norm_type is not None:
        if norm_type == "GN":
            layers.append(nn.GroupNorm(num_groups_for_gn, out_channels))
        elif norm_type == "LN":
            # channel-wise LN for conv maps handled by GroupNorm(1, C)
            layers.append(nn.GroupNorm(1, out_channels))
        else:
            layers.append(build_normalization(norm_type, dim=2)(out_channels))
    layers.append(build_activation(activation, inplace=False))
    return sequential_pack(layers)


def deconv2d_block(
    in_channels: int,
    out_channels: int,
    kernel_size: int,
    stride: int = 1,
    padding: int = 0,
    output_padding: int = 0,
    groups: int = 1,
    activation: Union[str, nn.Module, None] = None,
    norm_type: Optional[str] = None,
) -> nn.Sequential:
    layers = [
        nn.ConvTranspose2d(in_channels, out_channels, kernel_size, stride, padding, output_padding, groups)
    ]
    if norm_type is not None:
        layers.append(build_normalization(norm_type, dim=2)(out_channels))
    layers.append(build_activation(activation, inplace=False))
    return sequential_pack(layers)


def conv1d_block(
    in_channels: int,
    out_channels: int,
    kernel_size: int,
    stride: int = 1,
    padding: int = 0,
    dilation: int = 1,
    groups: int = 1,
    activation: Union[str, nn.Module, None] = None,
    norm_type: Optional[str] = None,
) -> nn.Sequential:
    layers = [nn.Conv1d(in_channels, out_channels, kernel_size, stride, padding, dilation, groups)]
    if norm_type is not None:
        layers.append(build_normalization(norm_type, dim=1)(out_channels))
    layers.append(build_activation(activation, inplace=False))
    return sequential_pack(layers)


def MLP(
    in_channels: int,
    hidden_channels: int,
    out_channels: int,
    layer_num: int,
    layer_fn: Callable = None,
    activation: Union[str, nn.Module, None] = None,
    norm_type: Optional[str] = None,
    use_dropout: bool = False,
    dropout_probability: float = 0.5,
    output_activation: bool = True,
    output_norm: bool = True,
    last_linear_layer_init_zero: bool = False,
) -> nn.Sequential:
    """layer_num linear layers in->hidden->...->out, with norm/activation
    after each (optionally excluding the last)."""
    assert layer_num >= 1
    channels = [in_channels] + [hidden_channels] * (layer_num - 1) + [out_channels]
    if layer_fn is None:
        layer_fn = nn.Linear
    layers: List[nn.Module] = []
    for i, (c_in, c_out) in enumerate(zip(channels[:-1], channels[1:])):
        last = i == layer_num - 1
        layers.append(layer_fn(c_in, c_out))
        if (not last) or output_norm:
            if norm_type is not None:
                layers.append(build_normalization(norm_type, dim=1)(c_out))
        if (not last) or output_activation:
            act = build_activation(activation, inplace=False)
            if act is not None:
                layers.append(act)
        if use_dropout and not last:
            layers.append(nn.Dropout(dropout_probability))
    if last_linear_layer_init_zero:
        for m in reversed(layers):
            if isinstance(m, nn.Linear):
                nn.init.zeros_(m.weight)
                nn.init.zeros_(m.bias)
                break
    return sequential_pack(layers)


def normed_linear(in_features: int, out_features: int, bias: bool = True, scale: float = 1.0) -> nn.Linear:
    """Linear layer with row-normalized init scaled by ``scale`` (IMPALA)."""
    layer = nn.Linear(in_features, out_features, bias=bias)
    with torch.no_grad():
        layer.weight.data *= scale / layer.weight.norm(dim=1, p=2, keepdim=True)
        if bias:
            layer.bias.data.zero_()
    return layer


def normed_conv2d(
    in_channels: int, out_channels: int, kernel_size: int, stride: int = 1, padding: int = 0, scale: float = 1.0
) -> nn.Conv2d:
    layer = nn.Conv2d(in_channels, out_channels, kernel_size, stride, padding)
    with torch.no_grad():
        layer.weight.data *= scale / layer.weight.norm(dim=(1, 2, 3), p=2, keepdim=True)
        layer.bias.data.zero_()
    return layer


def one_hot(val: torch.LongTensor, num: int, num_first: bool = False) -> torch.FloatTensor:
    """One-hot with -1 treated as all-zero row (padding-safe)."""
    assert isinstance(val, torch.Tensor)
    old_shape = val.shape
    val_flat = val.reshape(-1, 1)
    mask = (val_flat == -1)
    idx = val_flat.clamp(min=0)
    ret = torch.zeros(val_flat.shape[0], num, device=val.device, dtype=torch.float32)
    ret.scatter_(1, idx, 1.0)
    ret = ret * (~mask).float()
    if num_first:
        return ret.permute(1, 0).reshape(num, *old_shape)
    return ret.reshape(*old_shape, num)


def binary_encode(y: torch.Tensor, max_val: torch.Tensor) -> torch.Tensor:
    """Binary encoding of non-negative ints up to max_val."""
    if isinstance(max_val, torch.Tensor):
        max_val = int(max_val.item())
    L = int(max_val).bit_length()
    bits = torch.arange(L - 1, -1, -1, device=y.device)
    return ((y.unsqueeze(-1) >> bits) & 1).float()


class NoisyLinearLayer(nn.Module):
    """Factorized-Gaussian noisy linear (NoisyNet, Rainbow)."""

    def __init__(self, in_channels: int, out_channels: int, sigma0: float = 0.4):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.weight_mu = nn.Parameter(torch.empty(out_channels, in_channels))
        self.weight_sigma = nn.Parameter(torch.empty(out_channels, in_channels))
        self.bias_mu = nn.Parameter(torch.empty(out_channels))
        self.bias_sigma = nn.Parameter(torch.empty(out_channels))
        self.register_buffer("weight_eps", torch.zeros(out_channels, in_channels))
        self.register_buffer("bias_eps", torch.zeros(out_channels))
        self.sigma0 = sigma0
        self.reset_parameters()
        self.reset_noise()

    @staticmethod
    def _f(x: torch.Tensor) -> torch.Tensor:
        return torch.sign(x) * torch.sqrt(torch.abs(x))

    def reset_parameters(self):
        bound = 1 / math.sqrt(self.in_channels)
        for mu in (self.weight_mu, self.bias_mu):
            nn.init.uniform_(mu, -bound, bound)
        sigma_init = self.sigma0 / math.sqrt(self.in_channels)
        nn.init.constant_(self.weight_sigma, sigma_init)
        nn.init.constant_(self.bias_sigma, sigma_init)

    def reset_noise(self):
        device = self.weight_mu.device
        eps_in = self._f(torch.randn(self.in_channels, device=device))
        eps_out = self._f(torch.randn(self.out_channels, device=device))
        self.weight_eps = eps_out.outer(eps_in)
        self.bias_eps = eps_out

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.training:
            return F.linear(
                x, self.weight_mu + self.weight_sigma * self.weight_eps, self.bias_mu + self.bias_sigma * self.bias_eps
            )
        return F.linear(x, self.weight_mu, self.bias_mu)


def noise_block(
    in_channels: int,
    out_channels: int,
    activation: Union[str, nn.Module, None] = None,
    norm_type: Optional[str] = None,
    use_dropout: bool = False,
    dropout_probability: float = 0.5,
    sigma0: float = 0.4,
):
    layers = [NoisyLinearLayer(in_channels, out_channels, sigma0=sigma0)]
    if norm_type is not None:
        layers.append(build_normalization(norm_type, dim=1)(out_channels))
    layers.append(build_activation(activation))
    if use_dropout:
        layers.append(nn.Dropout(dropout_probability))
    return sequential_pack(layers)
