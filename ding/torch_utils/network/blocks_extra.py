"""Extra network blocks completing the reference's nn toolbox: activations,
upsamples, soft-argmax, Gumbel-Softmax, the AlphaStar-style stream mergers
and a compact ResNet.

Parity: reference ding/torch_utils/network/activation.py (Swish:81,
GumbelSoftmax via gumbel_softmax.py:6), nn_module.py (NearestUpsample:542,
BilinearUpsample:572, Flatten), soft_argmax.py (SoftArgmax:6), merge.py
(GatingType:220, SumMerge:230, VectorMerge:257), resnet.py (ResNet,
resnet18). The ResNet here is a compact channels-last-friendly BasicBlock
stack rather than a timm port: on MI355X the convs route to MIOpen NHWC
solvers, so the model only has to express the topology.
"""
import enum
from collections import OrderedDict
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


class Swish(nn.Module):
    """x * sigmoid(x) (SiLU)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x * torch.sigmoid(x)


class Flatten(nn.Module):
    """Flatten all dims after the batch dim (kept for reference parity;
    torch.nn.Flatten is equivalent)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x.reshape(x.shape[0], -1)


class NearestUpsample(nn.Module):

    def __init__(self, scale_factor) -> None:
        super().__init__()
        self.scale_factor = scale_factor

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.interpolate(x, scale_factor=self.scale_factor, mode='nearest')


class BilinearUpsample(nn.Module):

    def __init__(self, scale_factor) -> None:
        super().__init__()
        self.scale_factor = scale_factor

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.interpolate(x, scale_factor=self.scale_factor, mode='bilinear', align_corners=False)


class SoftArgmax(nn.Module):
    """Differentiable argmax over a [B, C, H, W] heatmap -> [B, 2] (h, w)
    expected coordinates."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape
        probs = F.softmax(x.reshape(B, C, -1), dim=-1).reshape(B, C, H, W)
        h_idx = torch.arange(H, dtype=x.dtype, device=x.device).view(1, 1, H, 1)
        w_idx = torch.arange(W, dtype=x.dtype, device=x.device).view(1, 1, 1, W)
        h = (probs * h_idx).sum(dim=(1, 2, 3))
        w = (probs * w_idx).sum(dim=(1, 2, 3))
        return torch.stack([h, w], dim=-1)


class GumbelSoftmax(nn.Module):
    """Gumbel-Softmax reparameterized categorical sampling; ``hard=True``
    returns straight-through one-hots."""

    def forward(self, x: torch.Tensor, temperature: float = 1.0, hard: bool = False) -> torch.Tensor:
        return F.gumbel_softmax(x, tau=temperature, hard=hard, dim=-1)

    def gumbel_softmax_sample(self, x: torch.Tensor, temperature: float = 1.0) -> torch.Tensor:
        return self.forward(x, temperature=temperature, hard=False)


class GatingType(enum.Enum):
    """How VectorMerge weighs its input streams."""
    NONE = 'none'
    GLOBAL = 'global'
    POINTWISE = 'pointwise'


class SumMerge(nn.Module):
    """Sum same-shaped streams."""

    def forward(self, tensors: List[torch.Tensor]) -> torch.Tensor:
        out = tensors[0]
        for t in tensors[1:]:
            out = out + t
        return out


class VectorMerge(nn.Module):
    """LayerNorm -> relu -> linear per stream, then a (optionally gated) sum.
    Streams may have different sizes; size<=0 marks a scalar stream."""

    def __init__(self, input_sizes: Dict[str, int], output_size: int,
                 gating_type: GatingType = GatingType.NONE, use_layer_norm: bool = True) -> None:
        super().__init__()
        self._input_sizes = OrderedDict(input_sizes)
        self._output_size = output_size
        self._gating_type = gating_type
        self._use_layer_norm = use_layer_norm
        self._layer_norms = nn.ModuleDict() if use_layer_norm else None
        self._linears = nn.ModuleDict()
        for name, size in self._input_sizes.items():
            in_size = size if (size or 0) > 0 else 1
            if use_layer_norm:
                self._layer_norms[name] = nn.LayerNorm(in_size)
            self._linears[name] = nn.Linear(in_size, output_size)
        if gating_type is GatingType.NONE:
            self._gating_linears = None
        else:
            self.gate_size = 1 if gating_type is GatingType.GLOBAL else output_size
            self._gating_linears = nn.ModuleDict()
            # two streams: a single shared sigmoid gate; more: per-stream softmax
            fan_out = self.gate_size if len(self._input_sizes) == 2 else len(self._input_sizes) * self.gate_size
            for name, size in self._input_sizes.items():
                lin = nn.Linear(size if (size or 0) > 0 else 1, fan_out)
                nn.init.normal_(lin.weight, std=0.005)
                nn.init.zeros_(lin.bias)
                self._gating_linears[name] = lin

    def encode(self, inputs: Dict[str, torch.Tensor]):
        gates, outputs = [], []
        for name, size in self._input_sizes.items():
            feat = inputs[name]
            if (size or 0) <= 0 and feat.dim() == 1:
                feat = feat.unsqueeze(-1)
            feat = feat.float()
            if self._use_layer_norm:
                feat = self._layer_norms[name](feat)
            feat = F.relu(feat)
            gates.append(feat)
            outputs.append(self._linears[name](feat))
        return gates, outputs

    def _compute_gate(self, init_gate: List[torch.Tensor]) -> List[torch.Tensor]:
        raw = [self._gating_linears[name](g) for name, g in zip(self._input_sizes, init_gate)]
        total = raw[0]
        for r in raw[1:]:
            total = total + r
        if len(self._input_sizes) == 2:
            sig = torch.sigmoid(total)
            return [sig, 1.0 - sig]
        total = total.reshape(-1, len(self._input_sizes), self.gate_size)
        soft = F.softmax(total, dim=1)
        return [soft[:, i] for i in range(len(self._input_sizes))]

    def forward(self, inputs: Dict[str, torch.Tensor]) -> torch.Tensor:
        gates, outputs = self.encode(inputs)
        if len(outputs) == 1:
            return outputs[0]
        if self._gating_type is GatingType.NONE:
            out = outputs[0]
            for o in outputs[1:]:
                out = out + o
            return out
        gate = self._compute_gate(gates)
        out = gate[0] * outputs[0]
        for g, o in zip(gate[1:], outputs[1:]):
            out = out + g * o
        return out


class _BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch: int, out_ch: int, stride: int = 1) -> None:
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, stride=1, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(out_ch)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False), nn.BatchNorm2d(out_ch))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.down is None else self.down(x)
        out = F.relu(self.bn1(self.conv1(x)), inplace=True)
        out = self.bn2(self.conv2(out))
        return F.relu(out + identity, inplace=True)


class ResNet(nn.Module):
    """Compact ResNet classifier: stem conv + 4 BasicBlock stages + GAP head."""

    def __init__(self, block=_BasicBlock, layers=(2, 2, 2, 2), num_classes: int = 1000,
                 in_chans: int = 3, base_width: int = 64) -> None:
        super().__init__()
        self.conv1 = nn.Conv2d(in_chans, base_width, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(base_width)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        chans = [base_width, base_width * 2, base_width * 4, base_width * 8]
        stages = []
        in_ch = base_width
        for i, (ch, n) in enumerate(zip(chans, layers)):
            blocks = []
            for j in range(n):
                blocks.append(block(in_ch, ch, stride=(2 if (i > 0 and j == 0) else 1)))
                in_ch = ch * block.expansion
            stages.append(nn.Sequential(*blocks))
        self.layer1, self.layer2, self.layer3, self.layer4 = stages
        self.num_features = in_ch
        self.fc = nn.Linear(in_ch, num_classes)

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.bn1(self.conv1(x)), inplace=True)
        x = self.maxpool(x)
        for stage in (self.layer1, self.layer2, self.layer3, self.layer4):
            x = stage(x)
        return x

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.forward_features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def resnet18(num_classes: int = 1000, in_chans: int = 3) -> ResNet:
    return ResNet(layers=(2, 2, 2, 2), num_classes=num_classes, in_chans=in_chans)
