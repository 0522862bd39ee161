"""Optimizers with integrated gradient clip/ignore, and PCGrad.

Parity: reference ding/torch_utils/optimizer_helper.py (Adam:107,
RMSprop:395, PCGrad:650, grad monitoring helpers).

Clip semantics (matching the reference's option set):
  grad_clip_type in {None, 'clip_value', 'clip_norm', 'clip_momentum_norm',
  'ignore_value', 'ignore_norm'} — clip_* rescales gradients, ignore_*
  zeroes the whole step when the threshold trips.
"""
import math
from typing import Iterable, List, Optional, Union

import torch
import torch.nn as nn

inf = math.inf


def calculate_grad_norm(model: torch.nn.Module, norm_type: int = 2) -> float:
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    if not grads:
        return 0.0
    return torch.norm(torch.stack([torch.norm(g.detach(), norm_type) for g in grads]), norm_type).item()


def calculate_grad_norm_without_bias_two_norm(model: torch.nn.Module) -> float:
    grads = [p.grad for n, p in model.named_parameters() if p.grad is not None and "bias" not in n]
    if not grads:
        return 0.0
    return torch.norm(torch.stack([torch.norm(g.detach(), 2) for g in grads]), 2).item()


def grad_ignore_norm(parameters, max_norm: float, norm_type: float = 2.0) -> float:
    """Zero all grads when total norm exceeds max_norm."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    parameters = [p for p in parameters if p.grad is not None]
    if not parameters:
        return 0.0
    total = torch.norm(torch.stack([torch.norm(p.grad.detach(), norm_type) for p in parameters]), norm_type)
    if total > max_norm:
        for p in parameters:
            p.grad.zero_()
    return float(total)


def grad_ignore_value(parameters, clip_value: float) -> None:
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    parameters = [p for p in parameters if p.grad is not None]
    flag = any(p.grad.detach().abs().max() > clip_value for p in parameters)
    if flag:
        for p in parameters:
            p.grad.zero_()


class Adam(torch.optim.Adam):
    """torch Adam + built-in gradient clip/ignore executed in step()."""

    def __init__(
        self,
        params: Iterable,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0,
        amsgrad: bool = False,
        optim_type: str = 'adam',
        grad_clip_type: Optional[str] = None,
        clip_value: Optional[float] = None,
        clip_coef: float = 5,
        clip_norm_type: float = 2.0,
        clip_momentum_timestep: int = 100,
        ignore_value: Optional[float] = None,
        ignore_coef: float = 5,
        ignore_norm_type: float = 2.0,
        ignore_momentum_timestep: int = 100,
        capturable: bool = False,
        flatten_grads: bool = False,
    ):
        self._grad_clip_type = grad_clip_type
        self._clip_value = clip_value
        self._clip_norm_type = clip_norm_type
        self._clip_coef = clip_coef
        self._ignore_value = ignore_value
        self._ignore_norm_type = ignore_norm_type
        self._ignore_coef = ignore_coef
        self._clip_momentum_timestep = clip_momentum_timestep
        self._ignore_momentum_timestep = ignore_momentum_timestep
        # capturable=True keeps Adam's step counters on-device so the whole
        # optimizer update can live inside a hipGraph capture (MI355X:
        # removes ~30 launches + host work per replayed minibatch)
        params = list(params)
        leaves = [p for p in params if isinstance(p, torch.Tensor)]
        # fused Adam: ONE multi-tensor kernel per step instead of the foreach
        # chain — rocprof on the PPO bench showed the eager optimizer
        # (multi_tensor_apply + fills + adds) at ~27 ms/step of the 180 ms
        # step. GPU-resident float params only; DING_FUSED_ADAM=0 disables.
        import os as _os
        use_fused = (
            _os.environ.get('DING_FUSED_ADAM', '1') not in ('0', 'false') and not capturable
            and len(leaves) == len(params) and len(leaves) > 0
            and all(p.is_cuda and p.dtype in (torch.float32, torch.float16, torch.bfloat16) for p in leaves)
        )
        super().__init__(
            params, lr=lr, betas=betas, eps=eps, weight_decay=weight_decay, amsgrad=amsgrad,
            capturable=capturable, foreach=(True if capturable else None) if not use_fused else None,
            fused=use_fused or None,
        )
        if grad_clip_type in ('clip_momentum_norm', 'ignore_momentum_norm'):
            for group in self.param_groups:
                group.setdefault('step_count', 0)
                group['grad_norm_ema'] = None
        # flat gradient buffer: every param.grad becomes a view into ONE
        # contiguous tensor, so grad-clip is 1 norm + 1 scale kernel and
        # zero_grad is 1 fill — rocprof showed the per-param reduce/fill
        # storm at ~25 ms/step on the PPO bench. Views have stable addresses,
        # which also makes backward hipGraph-capture friendly.
        self._flat_grad_buf = None
        if flatten_grads and leaves and all(p.is_cuda for p in leaves):
            total = sum(p.numel() for p in leaves)
            buf = torch.zeros(total, device=leaves[0].device, dtype=leaves[0].dtype)
            off = 0
            for p in leaves:
                p.grad = buf[off:off + p.numel()].view_as(p)
                off += p.numel()
            self._flat_grad_buf = buf

    def _params(self) -> List[torch.Tensor]:
        return [p for group in self.param_groups for p in group['params']]

    def _apply_clip(self):
        t = self._grad_clip_type
        if t is None:
            return
        params = self._params()
        if t == 'clip_norm' and self._flat_grad_buf is not None:
            buf = self._flat_grad_buf
            total_norm = torch.linalg.vector_norm(buf, self._clip_norm_type)
            scale = (self._clip_value / (total_norm + 1e-6)).clamp(max=1.0)
            buf.mul_(scale)
            return
        if t == 'clip_value':
            nn.utils.clip_grad_value_(params, self._clip_value)
        elif t == 'clip_norm':
            # foreach=True: one fused norm + scale instead of a reduce per param
            _fe = True if (params and all(p.is_cuda for p in params)) else None
            nn.utils.clip_grad_norm_(params, self._clip_value, self._clip_norm_type, foreach=_fe)
        elif t == 'ignore_value':
            grad_ignore_value(params, self._ignore_value)
        elif t == 'ignore_norm':
            grad_ignore_norm(params, self._ignore_value, self._ignore_norm_type)
        elif t in ('clip_momentum_norm', 'ignore_momentum_norm'):
            # threshold = coef * EMA of historical grad norms
            for group in self.param_groups:
                ps = [p for p in group['params'] if p.grad is not None]
                if not ps:
                    continue
                norm = torch.norm(
                    torch.stack([torch.norm(p.grad.detach(), self._clip_norm_type) for p in ps]),
                    self._clip_norm_type
                ).item()
                ema = group.get('grad_norm_ema')
                ema = norm if ema is None else 0.99 * ema + 0.01 * norm
                group['grad_norm_ema'] = ema
                group['step_count'] = group.get('step_count', 0) + 1
                if group['step_count'] >= self._clip_momentum_timestep:
                    coef = self._clip_coef if t == 'clip_momentum_norm' else self._ignore_coef
                    threshold = coef * ema
                    if norm > threshold:
                        if t == 'clip_momentum_norm':
                            scale = threshold / (norm + 1e-6)
                            for p in ps:
                                p.grad.mul_(scale)
                        else:
                            for p in ps:
                                p.grad.zero_()
        else:
            raise KeyError(f"unknown grad_clip_type {t}")

    def step(self, closure=None):
        self._apply_clip()
        return super().step(closure)

    def zero_grad(self, set_to_none: bool = True):
        if self._flat_grad_buf is not None:
            # keep the grad views alive: one fill over the flat buffer
            self._flat_grad_buf.zero_()
            return
        return super().zero_grad(set_to_none)

    def get_grad(self) -> float:
        return calculate_grad_norm_without_bias_two_norm(_FakeModel(self._params()))


class _FakeModel:

    def __init__(self, params):
        self._p = params

    def named_parameters(self):
        return [(f"p{i}", p) for i, p in enumerate(self._p)]

    def parameters(self):
        return self._p


class RMSprop(torch.optim.RMSprop):
    """torch RMSprop + built-in gradient clip/ignore."""

    def __init__(
        self,
        params: Iterable,
        lr: float = 1e-2,
        alpha: float = 0.99,
        eps: float = 1e-8,
        weight_decay: float = 0,
        momentum: float = 0,
        centered: bool = False,
        grad_clip_type: Optional[str] = None,
        clip_value: Optional[float] = None,
        clip_norm_type: float = 2.0,
        ignore_value: Optional[float] = None,
        ignore_norm_type: float = 2.0,
    ):
        self._grad_clip_type = grad_clip_type
        self._clip_value = clip_value
        self._clip_norm_type = clip_norm_type
        self._ignore_value = ignore_value
        self._ignore_norm_type = ignore_norm_type
        super().__init__(
            params, lr=lr, alpha=alpha, eps=eps, weight_decay=weight_decay, momentum=momentum, centered=centered
        )

    def _params(self):
        return [p for group in self.param_groups for p in group['params']]

    def step(self, closure=None):
        t = self._grad_clip_type
        params = self._params()
        if t == 'clip_value':
            nn.utils.clip_grad_value_(params, self._clip_value)
        elif t == 'clip_norm':
            # foreach=True: one fused norm + scale instead of a reduce per param
            _fe = True if (params and all(p.is_cuda for p in params)) else None
            nn.utils.clip_grad_norm_(params, self._clip_value, self._clip_norm_type, foreach=_fe)
        elif t == 'ignore_value':
            grad_ignore_value(params, self._ignore_value)
        elif t == 'ignore_norm':
            grad_ignore_norm(params, self._ignore_value, self._ignore_norm_type)
        return super().step(closure)


class PCGrad:
    """Projecting Conflicting Gradients (NeurIPS 2020) wrapper for multi-task
    losses. Parity: reference optimizer_helper.py:650."""

    def __init__(self, optimizer: torch.optim.Optimizer, reduction: str = 'mean'):
        self._optim = optimizer
        assert reduction in ('mean', 'sum')
        self._reduction = reduction

    @property
    def optimizer(self):
        return self._optim

    def zero_grad(self):
        self._optim.zero_grad(set_to_none=False)

    def step(self):
        self._optim.step()

    def pc_backward(self, objectives: List[torch.Tensor]):
        grads, shapes, has_grads = [], [], []
        for obj in objectives:
            self._optim.zero_grad(set_to_none=False)
            obj.backward(retain_graph=True)
            grad, shape, has_grad = self._retrieve_grad()
            grads.append(self._flatten(grad))
            has_grads.append(self._flatten(has_grad))
            shapes.append(shape)
        pc_grad = self._project_conflicting(grads, has_grads)
        pc_grad = self._unflatten(pc_grad, shapes[0])
        self._set_grad(pc_grad)

    def _project_conflicting(self, grads, has_grads):
        import copy
        import random
        shared = torch.stack(has_grads).prod(0).bool()
        pc_grad = [g.clone() for g in grads]
        for g_i in pc_grad:
            others = list(grads)
            random.shuffle(others)
            for g_j in others:
                dot = torch.dot(g_i, g_j)
                if dot < 0:
                    g_i -= dot * g_j / (g_j.norm() ** 2 + 1e-12)
        merged = torch.zeros_like(grads[0])
        stacked = torch.stack(pc_grad)
        if self._reduction == 'mean':
            merged[shared] = stacked[:, shared].mean(dim=0)
        else:
            merged[shared] = stacked[:, shared].sum(dim=0)
        merged[~shared] = stacked[:, ~shared].sum(dim=0)
        return merged

    def _retrieve_grad(self):
        grad, shape, has_grad = [], [], []
        for group in self._optim.param_groups:
            for p in group['params']:
                shape.append(p.shape)
                if p.grad is None:
                    grad.append(torch.zeros_like(p))
                    has_grad.append(torch.zeros_like(p))
                else:
                    grad.append(p.grad.clone())
                    has_grad.append(torch.ones_like(p))
        return grad, shape, has_grad

    @staticmethod
    def _flatten(grads):
        return torch.cat([g.reshape(-1) for g in grads])

    @staticmethod
    def _unflatten(flat, shapes):
        out, idx = [], 0
        for s in shapes:
            n = int(torch.tensor(s).prod().item()) if len(s) else 1
            out.append(flat[idx:idx + n].reshape(s))
            idx += n
        return out

    def _set_grad(self, grads):
        i = 0
        for group in self._optim.param_groups:
            for p in group['params']:
                p.grad = grads[i]
                i += 1


def configure_weight_decay(model: nn.Module, weight_decay: float) -> List[dict]:
    """Split params into decay (matmul weights) / no-decay (bias, norm, emb)."""
    decay, no_decay = [], []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        if p.dim() <= 1 or name.endswith("bias") or "norm" in name.lower() or "embedding" in name.lower():
            no_decay.append(p)
        else:
            decay.append(p)
    return [
        {"params": decay, "weight_decay": weight_decay},
        {"params": no_decay, "weight_decay": 0.0},
    ]
