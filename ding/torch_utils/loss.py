"""Loss modules: label-smoothed CE, soft-target CE, multi-logits matching
loss, InfoNCE contrastive loss.

Parity: reference ding/torch_utils/loss/ (cross_entropy_loss.py,
multi_logits_loss.py, contrastive_loss.py).
"""
from collections import namedtuple
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

ce_loss = namedtuple('ce_loss', ['loss'])


class LabelSmoothCELoss(nn.Module):

    def __init__(self, ratio: float):
        super().__init__()
        self.ratio = ratio

    def forward(self, logits: torch.Tensor, labels: torch.LongTensor) -> torch.Tensor:
        B, N = logits.shape
        val = self.ratio / N
        one_hot = torch.full_like(logits, val)
        one_hot.scatter_(1, labels.unsqueeze(1), 1 - self.ratio + val)
        logits = F.log_softmax(logits, dim=1)
        return -(logits * one_hot.detach()).sum(dim=1).mean()


class SoftLogitsLoss(nn.Module):
    """CE against a soft target distribution (distillation)."""

    def __init__(self, T: float = 1.0):
        super().__init__()
        self.T = T

    def forward(self, logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
        logp = F.log_softmax(logits / self.T, dim=1)
        target_p = F.softmax(targets / self.T, dim=1)
        return -(target_p * logp).sum(dim=1).mean()


def build_ce_criterion(cfg: dict) -> nn.Module:
    t = cfg.get('type', 'cross_entropy')
    if t == 'cross_entropy':
        return nn.CrossEntropyLoss()
    if t == 'label_smooth_ce':
        return LabelSmoothCELoss(cfg['kwargs']['smooth_ratio'])
    if t == 'soft_ce':
        return SoftLogitsLoss(cfg.get('kwargs', {}).get('T', 1.0))
    raise KeyError(t)


class MultiLogitsLoss(nn.Module):
    """CE between a set of logits heads and a set of labels where the
    head->label assignment is found greedily/by matching (AlphaStar selected
    units head). criterion over [H, N] logits vs label list."""

    def __init__(self, criterion: str = 'cross_entropy', smooth_ratio: float = 0.1):
        super().__init__()
        if criterion == 'cross_entropy':
            self._ce = lambda logit, label: -F.log_softmax(logit, dim=0)[label]
        elif criterion == 'label_smooth_ce':
            ratio = smooth_ratio

            def _smooth(logit, label):
                N = logit.shape[0]
                val = ratio / N
                one_hot = torch.full_like(logit, val)
                one_hot[label] = 1 - ratio + val
                return -(F.log_softmax(logit, dim=0) * one_hot).sum()

            self._ce = _smooth
        else:
            raise KeyError(criterion)

    def forward(self, logits: torch.Tensor, labels: torch.LongTensor) -> torch.Tensor:
        """logits [H, N], labels [M] with M <= H: greedy min-cost assignment."""
        H, N = logits.shape
        M = labels.shape[0]
        assert M <= H
        cost = torch.stack([torch.stack([self._ce(logits[h], labels[m]) for m in range(M)]) for h in range(H)])
        # greedy assignment by ascending cost
        with torch.no_grad():
            flat = cost.detach().reshape(-1)
            order = torch.argsort(flat)
            used_h, used_m, pairs = set(), set(), []
            for idx in order.tolist():
                h, m = idx // M, idx % M
                if h in used_h or m in used_m:
                    continue
                used_h.add(h)
                used_m.add(m)
                pairs.append((h, m))
                if len(pairs) == M:
                    break
        return torch.stack([cost[h, m] for h, m in pairs]).mean()


class ContrastiveLoss(nn.Module):
    """InfoNCE lower bound on MI(x, y) with a bilinear critic (ST-DIM)."""

    def __init__(
        self,
        x_size,
        y_size,
        heads: list = None,
        encode_shape: int = 64,
        loss_type: str = 'infonce',
        temperature: float = 1.0,
    ):
        super().__init__()
        self._encode_shape = encode_shape
        self._temperature = temperature
        self._x_encoder = self._get_encoder(x_size, encode_shape)
        self._y_encoder = self._get_encoder(y_size, encode_shape)
        self._W = nn.Parameter(torch.randn(encode_shape, encode_shape) * 0.02)

    @staticmethod
    def _get_encoder(obs_size, out_size: int) -> nn.Module:
        if isinstance(obs_size, int):
            return nn.Sequential(nn.Linear(obs_size, 128), nn.ReLU(), nn.Linear(128, out_size))
        # conv encoder for image obs [C, H, W]
        c = obs_size[0]
        return nn.Sequential(
            nn.Conv2d(c, 32, 8, 4), nn.ReLU(), nn.Conv2d(32, 64, 4, 2), nn.ReLU(), nn.Flatten(),
            nn.LazyLinear(out_size)
        )

    def forward(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        B = x.shape[0]
        ex = self._x_encoder(x.float())  # [B, E]
        ey = self._y_encoder(y.float())
        logits = ex @ self._W @ ey.t() / self._temperature  # [B, B]
        labels = torch.arange(B, device=x.device)
        return F.cross_entropy(logits, labels)
