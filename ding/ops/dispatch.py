"""Runtime dispatch between the pure-PyTorch oracle lane and the HIP kernels.

The HIP extension is the in-tree shared object ``ding/ops/_hiprl*.so`` built
for gfx950. On a machine WITH a GPU the extension is required: if a hot op
receives a CUDA(HIP) tensor and the extension is absent we raise — a silent
eager fallback would invalidate benchmarks. CPU tensors always use the
PyTorch lane (that lane doubles as the numerics oracle in tests).
"""
import os
from typing import Optional

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


def _load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from ding.ops import _hiprl  # built by setup_ops.py / __graft_entry__.build()
        _EXT = _hiprl
    except ImportError as e:
        _EXT_ERR = str(e)
    return _EXT


def is_available() -> bool:
    return _load() is not None


def hip_ops():
    """Return the raw extension module (None if unavailable)."""
    return _load()


_DISABLED = os.environ.get("DI_ENGINE_DISABLE_HIP", "0") in ("1", "true", "True")


def use_hip(x: torch.Tensor) -> bool:
    """Decide the lane for this call. Raises on GPU-without-extension.

    Forward-only kernels: if gradient would flow through ``x`` we fall back
    to the differentiable PyTorch lane (ops with analytic backwards expose
    their own autograd.Function entry points instead).
    """
    if _DISABLED or not isinstance(x, torch.Tensor) or not x.is_cuda:
        return False
    if torch.is_grad_enabled() and x.requires_grad:
        return False
    if not is_available():
        raise RuntimeError(
            f"ding.ops HIP extension not built but got a GPU tensor (import error: {_EXT_ERR}). "
            "Run `python setup_ops.py` (or __graft_entry__.build()) to compile for gfx950, "
            "or set DI_ENGINE_DISABLE_HIP=1 to force the PyTorch lane."
        )
    return True


# ----------------------------------------------------------------- wrappers
# Each wrapper matches the call contract documented in its rl_utils caller.

def gae_scan(delta: torch.Tensor, factor) -> torch.Tensor:
    """adv[t] = delta[t] + factor[t] * adv[t+1] reverse scan over dim 0.

    factor may be a tensor shaped like delta or a python float.
    """
    ext = _load()
    delta2d = delta.reshape(delta.shape[0], -1)
    if isinstance(factor, torch.Tensor):
        factor2d = factor.expand_as(delta).reshape(delta.shape[0], -1).contiguous()
    else:
        factor2d = torch.full_like(delta2d, float(factor))
    out = ext.reverse_scan(delta2d.contiguous().float(), factor2d.float())
    return out.reshape(delta.shape).to(delta.dtype)


def multistep_forward_view(bootstrap_values, rewards, gammas, lambda_, done):
    ext = _load()
    return ext.multistep_forward_view(
        bootstrap_values.contiguous().float(), rewards.contiguous().float(), gammas.contiguous().float(),
        lambda_.contiguous().float(), done.contiguous().float()
    ).to(rewards.dtype)


def scatter_connection(x, index, H: int, W: int, scatter_type: str):
    """x [B,M,N], index [B,M] flat spatial positions -> [B,N,H,W]."""
    ext = _load()
    return ext.scatter_connection(
        x.contiguous().float(), index.contiguous().long(), int(H), int(W), 1 if scatter_type == 'add' else 0
    ).to(x.dtype)


def c51_project(next_n_dist, next_n_act, reward_n, done, v_min, v_max, gamma_n):
    """Categorical projection of r + gamma_n * z onto the fixed support.

    next_n_dist [B,N,n_atom], next_n_act [B], reward_n/done [B] ->
    proj [B, n_atom]. Grad-free (target side of the C51 loss).
    """
    ext = _load()
    return ext.c51_project(
        next_n_dist.contiguous().float(), next_n_act.contiguous().long(), reward_n.contiguous().float(),
        done.contiguous().float(), float(v_min), float(v_max), float(gamma_n)
    )
