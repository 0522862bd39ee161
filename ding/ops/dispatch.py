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


def use_hip_autograd(x: torch.Tensor) -> bool:
    """Lane check for fused ops that implement their own backward (gradient
    through ``x`` is fine)."""
    if _DISABLED or not isinstance(x, torch.Tensor) or not x.is_cuda:
        return False
    if not is_available():
        raise RuntimeError(
            f"ding.ops HIP extension not built but got a GPU tensor (import error: {_EXT_ERR}). "
            "Run `python setup_ops.py` or set DI_ENGINE_DISABLE_HIP=1."
        )
    return True


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


class _FusedPPODiscrete(torch.autograd.Function):
    """Fused discrete PPO loss: forward packs the per-row terms in one kernel,
    backward produces analytic d(logit_new)/d(value_new) in one kernel."""

    @staticmethod
    def forward(ctx, logit_new, value_new, logit_old, action, value_old, adv, ret, weight, clip_ratio,
                use_value_clip):
        ext = _load()
        w = weight if weight is not None else torch.empty(0, device=logit_new.device)
        out = ext.ppo_fwd(
            logit_new.contiguous(), logit_old.contiguous(), action.contiguous(), value_new.contiguous(),
            value_old.contiguous(), adv.contiguous(), ret.contiguous(), w, float(clip_ratio), int(use_value_clip)
        )[0]
        ctx.save_for_backward(logit_new, action, value_new, value_old, adv, ret, w, out)
        ctx.clip_ratio = float(clip_ratio)
        policy_loss = out[:, 0].mean()
        value_loss = out[:, 1].mean()
        entropy_loss = out[:, 2].mean()
        approx_kl = out[:, 3].mean()
        clipfrac = out[:, 4].mean()
        ctx.mark_non_differentiable(approx_kl, clipfrac)
        return policy_loss, value_loss, entropy_loss, approx_kl, clipfrac

    @staticmethod
    def backward(ctx, g_policy, g_value, g_entropy, g_kl=None, g_clip=None):
        ext = _load()
        logit_new, action, value_new, value_old, adv, ret, w, out = ctx.saved_tensors
        # upstream grad scales stay DEVICE-resident ([3] tensor) so the
        # backward launch is hipGraph-capture safe and sync-free
        gs = torch.stack([
            g_policy.reshape(()), g_value.reshape(()), g_entropy.reshape(())
        ]).float().contiguous()
        d_logit, d_value = ext.ppo_bwd(
            logit_new.contiguous(), action.contiguous(), value_new.contiguous(), value_old.contiguous(),
            adv.contiguous(), ret.contiguous(), w, out, ctx.clip_ratio, gs
        )
        return d_logit, d_value, None, None, None, None, None, None, None, None


def fused_ppo_error(logit_new, logit_old, action, value_new, value_old, adv, ret, weight,
                    clip_ratio: float, use_value_clip: bool):
    """Returns (policy_loss, value_loss, entropy_loss, approx_kl, clipfrac)."""
    return _FusedPPODiscrete.apply(
        logit_new, value_new, logit_old, action, value_old, adv, ret, weight, clip_ratio, use_value_clip
    )


class _FusedVtraceDiscrete(torch.autograd.Function):
    """Fully fused discrete v-trace loss: 2 forward launches (row log-softmax/
    IS/entropy + column scan with vs/adv/loss contribs), 1 analytic backward
    launch. Parity: reference ding/rl_utils/vtrace.py:73."""

    @staticmethod
    def forward(ctx, t_logit, value, b_logit, action, reward, weight, gamma, lambda_, rho_c, c_c, rho_pg_c):
        ext = _load()
        w = weight if weight is not None else torch.empty(0, device=t_logit.device)
        row_out, ret, adv, contrib = ext.vtrace_fwd(
            t_logit.contiguous(), b_logit.contiguous(), action.contiguous().long(), value.contiguous(),
            reward.contiguous(), w, float(gamma), float(lambda_), float(rho_c), float(c_c), float(rho_pg_c)
        )
        ctx.save_for_backward(t_logit, row_out, value, ret, adv, action, w)
        losses = contrib.mean(dim=0)  # [3]: pg, value, entropy
        return losses[0], losses[1], losses[2]

    @staticmethod
    def backward(ctx, g_pg, g_v, g_ent):
        ext = _load()
        t_logit, row_out, value, ret, adv, action, w = ctx.saved_tensors
        # device-resident [3] so the launch is hipGraph-capture safe
        gs = torch.stack([g_pg.reshape(()), g_v.reshape(()), g_ent.reshape(())]).float().contiguous()
        d_logit, d_value = ext.vtrace_bwd(
            t_logit.contiguous(), row_out, value.contiguous(), ret, adv, action.contiguous().long(), w, gs
        )
        return d_logit, d_value, None, None, None, None, None, None, None, None, None


def fused_vtrace_error(t_logit, b_logit, action, value, reward, weight, gamma, lambda_, rho_c, c_c, rho_pg_c):
    """Returns (policy_loss, value_loss, entropy_loss)."""
    return _FusedVtraceDiscrete.apply(
        t_logit, value, b_logit, action, reward, weight, gamma, lambda_, rho_c, c_c, rho_pg_c
    )


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


class _FusedQNstepTD(torch.autograd.Function):
    """Fused n-step Q TD: HIP forward computes {td, return, q_sa} in one
    launch; backward is the single scatter d(td)/d(q[b,a]) = 1 (target
    detached), expressed in eager torch."""

    @staticmethod
    def forward(ctx, q, next_n_q, action, next_action, reward, done, value_gamma, gamma, nstep, rescale):
        ext = _load()
        vg = value_gamma if value_gamma is not None else torch.empty(0, device=q.device)
        (out, ) = ext.q_nstep_fwd(
            q.detach().contiguous(), next_n_q.detach().contiguous(), action.contiguous(),
            next_action.contiguous(), reward.detach().contiguous(), done.detach().float().contiguous(), vg,
            float(gamma), int(nstep), int(rescale)
        )
        td, ret, q_sa = out[:, 0], out[:, 1], out[:, 2]
        ctx.save_for_backward(action)
        ctx.q_shape = q.shape
        ctx.mark_non_differentiable(ret)
        return td, ret

    @staticmethod
    def backward(ctx, g_td, g_ret=None):
        (action, ) = ctx.saved_tensors
        d_q = torch.zeros(ctx.q_shape, device=g_td.device, dtype=g_td.dtype)
        d_q.scatter_(1, action.unsqueeze(1), g_td.unsqueeze(1))
        return d_q, None, None, None, None, None, None, None, None, None


def fused_q_nstep_td(q, next_n_q, action, next_action, reward, done, value_gamma, gamma, nstep,
                     rescale: bool = False):
    """Returns (td = q_sa - return  [differentiable wrt q], return)."""
    return _FusedQNstepTD.apply(q, next_n_q, action, next_action, reward, done, value_gamma, gamma, nstep,
                                1 if rescale else 0)


class _StemConvFn(torch.autograd.Function):
    """Direct 8x8s4 stem conv: HIP forward + HIP weight-grad; no input grad
    (input layer). Bias grad is a cheap eager reduction."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        ext = _load()
        y = ext.stem_conv_fwd(
            x.detach().contiguous(), weight.detach().contiguous(),
            bias.detach().contiguous() if bias is not None else torch.empty(0, device=x.device)
        )
        ctx.save_for_backward(x)
        ctx.O = weight.shape[0]
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        (x, ) = ctx.saved_tensors
        dy = dy.contiguous()
        # dW as ONE hipBLASLt GEMM: dW[O, C*64] = dY'[O, B*P] @ cols[B*P, C*64]
        # (the HIP wrw kernel's per-entry gather measured 3.7 ms/call — the
        # uncoalesced dW-major loop loses to im2col+GEMM here)
        B, O = dy.shape[0], dy.shape[1]
        cols = torch.nn.functional.unfold(x, 8, stride=4)          # [B, C*64, P]
        dyf = dy.reshape(B, O, -1)                                 # [B, O, P]
        dw = torch.bmm(dyf, cols.transpose(1, 2)).sum(0)           # [O, C*64]
        dw = dw.reshape(O, x.shape[1], 8, 8)
        db = dy.sum(dim=(0, 2, 3)) if ctx.has_bias else None
        return None, dw, db


def stem_conv2d(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor = None) -> torch.Tensor:
    """conv2d(x, w, b, stride=4) for 8x8 kernels, pad 0, W_out<=32."""
    return _StemConvFn.apply(x, weight, bias)


class _FusedLSTMCell(torch.autograd.Function):
    """One-launch LN-LSTM cell: LN(gh_raw) + gates + state update; backward
    is one kernel + eager per-parameter reductions. The gradient w.r.t.
    gxn (input-side gates) flows back into the eager sequence-level LN_x."""

    @staticmethod
    def forward(ctx, gxn, gh_raw, gamma, beta, bias, c_in):
        ext = _load()
        h_out, c_out, xhat, acts, rstd = ext.lstm_cell_fwd(
            gxn.detach().contiguous(), gh_raw.detach().contiguous(), gamma.detach().contiguous(),
            beta.detach().contiguous(), bias.detach().contiguous(), c_in.detach().contiguous()
        )
        ctx.save_for_backward(acts, xhat, gamma, c_in, c_out, rstd)
        return h_out, c_out

    @staticmethod
    def backward(ctx, dh, dc_next):
        ext = _load()
        acts, xhat, gamma, c_in, c_out, rstd = ctx.saved_tensors
        dg, dgh, dc_in = ext.lstm_cell_bwd(
            dh.contiguous(), dc_next.contiguous() if dc_next is not None else torch.empty(0, device=dh.device),
            acts, xhat, gamma, c_in.contiguous(), c_out, rstd
        )
        # per-parameter reductions over the batch (single eager sums)
        dgamma = (dg * xhat).sum(0)
        dbeta = dg.sum(0)
        dbias = dg.sum(0)
        return dg, dgh, dgamma, dbeta, dbias, dc_in


def fused_lstm_cell(gxn, gh_raw, gamma, beta, bias, c_in):
    """Returns (h', c')."""
    return _FusedLSTMCell.apply(gxn, gh_raw, gamma, beta, bias, c_in)


class _WrwConv2dFn(torch.autograd.Function):
    """conv2d with MIOpen fwd / bwd-data but the HIP NHWC wrw kernel for
    dW — MIOpen resolves NHWC fp32 wrw for the Atari shapes to a naive
    fp64-accumulate solver (profiles/impala_nhwc_kernel_stats_r01.csv)."""

    @staticmethod
    def forward(ctx, x, weight, bias, stride):
        y = torch.nn.functional.conv2d(x, weight, bias, stride)
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load()
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = torch.nn.grad.conv2d_input(x.shape, weight, dy, stride=ctx.stride)
        dw = ext.conv_wrw_nhwc(x, dy, weight.shape[-1], ctx.stride[0] if isinstance(ctx.stride, (tuple, list)) else ctx.stride)
        db = dy.sum(dim=(0, 2, 3)) if ctx.has_bias else None
        return dx, dw, db, None


def wrw_conv2d(x, weight, bias, stride):
    """conv2d(x, w, b, stride) with the hand-written NHWC wrw backward."""
    return _WrwConv2dFn.apply(x, weight, bias, stride)
