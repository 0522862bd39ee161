"""Temporal-difference losses (1-step / n-step / distributional / quantile /
lambda family).

Parity: reference ding/rl_utils/td.py (q_nstep_td_error:649,
dist_nstep_td_error:413, td_lambda_error:1539, generalized_lambda_returns:1574,
multistep_forward_view:1608 and friends). Implementations are written fresh;
hot [T,B,N]-shaped entry points dispatch to the HIP/CDNA4 kernels in
``ding.ops`` when inputs live on a GPU (see ding/ops/dispatch.py).
"""
from collections import namedtuple
from typing import Optional, Union

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from .value_rescale import value_transform, value_inv_transform

# ---------------------------------------------------------------- namedtuples
q_1step_td_data = namedtuple('q_1step_td_data', ['q', 'next_q', 'act', 'next_act', 'reward', 'done', 'weight'])
m_q_1step_td_data = namedtuple('m_q_1step_td_data', ['q', 'target_q', 'next_q', 'act', 'reward', 'done', 'weight'])
q_v_1step_td_data = namedtuple('q_v_1step_td_data', ['q', 'v', 'act', 'reward', 'done', 'weight'])
nstep_return_data = namedtuple('nstep_return_data', ['reward', 'next_value', 'done'])
dist_1step_td_data = namedtuple(
    'dist_1step_td_data', ['dist', 'next_dist', 'act', 'next_act', 'reward', 'done', 'weight']
)
dist_nstep_td_data = namedtuple(
    'dist_nstep_td_data', ['dist', 'next_n_dist', 'act', 'next_n_act', 'reward', 'done', 'weight']
)
v_1step_td_data = namedtuple('v_1step_td_data', ['v', 'next_v', 'reward', 'done', 'weight'])
v_nstep_td_data = namedtuple('v_nstep_td_data', ['v', 'next_n_v', 'reward', 'done', 'weight', 'value_gamma'])
q_nstep_td_data = namedtuple(
    'q_nstep_td_data', ['q', 'next_n_q', 'action', 'next_n_action', 'reward', 'done', 'weight']
)
dqfd_nstep_td_data = namedtuple(
    'dqfd_nstep_td_data', [
        'q', 'next_n_q', 'action', 'next_n_action', 'reward', 'done', 'done_one_step', 'weight', 'new_n_q_one_step',
        'next_n_action_one_step', 'is_expert'
    ]
)
qrdqn_nstep_td_data = namedtuple(
    'qrdqn_nstep_td_data', ['q', 'next_n_q', 'action', 'next_n_action', 'reward', 'done', 'tau', 'weight']
)
iqn_nstep_td_data = namedtuple(
    'iqn_nstep_td_data', ['q', 'next_n_q', 'action', 'next_n_action', 'reward', 'done', 'replay_quantiles', 'weight']
)
fqf_nstep_td_data = namedtuple(
    'fqf_nstep_td_data', ['q', 'next_n_q', 'action', 'next_n_action', 'reward', 'done', 'quantiles_hats', 'weight']
)
td_lambda_data = namedtuple('td_lambda_data', ['value', 'reward', 'weight'])


def discount_cumsum(x, gamma: float = 1.0) -> np.ndarray:
    """Reverse discounted cumulative sum along axis 0 (numpy)."""
    x = np.asarray(x, dtype=np.float64)
    out = np.zeros_like(x)
    acc = np.zeros_like(x[0] if x.ndim > 1 else np.float64(0.0))
    for t in range(x.shape[0] - 1, -1, -1):
        acc = x[t] + gamma * acc
        out[t] = acc
    return out


def view_similar(x: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Right-pad singleton dims on x so it broadcasts against target."""
    if isinstance(x, torch.Tensor):
        while x.dim() < target.dim():
            x = x.unsqueeze(-1)
    return x


# ------------------------------------------------------------------- returns
def nstep_return(data: namedtuple, gamma: Union[float, list], nstep: int, value_gamma: Optional[torch.Tensor] = None):
    """G = sum_{i<n} gamma^i r_i + gamma^n V(s_{t+n}) (1 - done).

    reward: [T(=nstep), B]; next_value, done: [B] (or broadcastable).
    list-typed gamma handles per-sample discounts (NGU).
    """
    reward, next_value, done = data
    assert reward.shape[0] == nstep
    device = reward.device
    if isinstance(gamma, float) or np.isscalar(gamma):
        factor = gamma ** torch.arange(nstep, dtype=reward.dtype, device=device)
        factor = view_similar(factor, reward)
        ret = (reward * factor).sum(0)
        if value_gamma is None:
            tail = (gamma ** nstep) * next_value * (1 - done.float())
        else:
            if np.isscalar(value_gamma):
                value_gamma = torch.full_like(next_value, value_gamma)
            value_gamma = view_similar(value_gamma, next_value)
            tail = value_gamma * next_value * (1 - view_similar(done.float(), next_value))
        return ret + tail
    elif isinstance(gamma, list):
        g = torch.stack([torch.as_tensor(x, dtype=reward.dtype, device=device) for x in gamma], dim=0) \
            if not isinstance(gamma[0], torch.Tensor) else torch.stack(gamma, dim=0).to(device)
        # per-sample gamma: factor[i] = g^i, shape [nstep+1, B]
        factor = torch.ones(nstep + 1, done.shape[0], dtype=reward.dtype, device=device)
        for i in range(1, nstep + 1):
            factor[i] = g * factor[i - 1]
        ret = (reward * view_similar(factor[:nstep], reward)).sum(0)
        return ret + factor[nstep] * next_value * (1 - done.float())
    raise TypeError(f"gamma must be float or list, got {type(gamma)}")


def generalized_lambda_returns(
    bootstrap_values: torch.Tensor,
    rewards: torch.Tensor,
    gammas,
    lambda_,
    done: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """TRFL-style lambda returns. bootstrap_values: [T+1, B]; rewards [T, B]."""
    if not isinstance(gammas, torch.Tensor):
        gammas = gammas * torch.ones_like(rewards)
    if not isinstance(lambda_, torch.Tensor):
        lambda_ = lambda_ * torch.ones_like(rewards)
    return multistep_forward_view(bootstrap_values[1:, :], rewards, gammas, lambda_, done)


def multistep_forward_view(
    bootstrap_values: torch.Tensor,
    rewards: torch.Tensor,
    gammas: torch.Tensor,
    lambda_: torch.Tensor,
    done: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Sutton & Barto (12.18) forward-view reverse recursion.

    result[T-1] = r[T-1] + (1-done[T-1]) g[T-1] V[T]
    result[t]   = r[t]   + (1-done[t]) (g l result[t+1] + g(1-l) V[t+1])
    bootstrap_values here is V at steps 1..T, shape [T, B].
    """
    from ding.ops import dispatch
    if done is None:
        done = torch.zeros_like(rewards)
    if dispatch.use_hip(rewards):
        return dispatch.multistep_forward_view(bootstrap_values, rewards, gammas, lambda_, done)
    result = torch.empty_like(rewards)
    T = rewards.shape[0]
    result[T - 1] = rewards[T - 1] + (1 - done[T - 1]) * gammas[T - 1] * bootstrap_values[T - 1]
    disc = gammas * lambda_
    for t in range(T - 2, -1, -1):
        result[t] = rewards[t] + (1 - done[t]) * (
            disc[t] * result[t + 1] + (gammas[t] - disc[t]) * bootstrap_values[t]
        )
    return result


def td_lambda_error(data: namedtuple, gamma: float = 0.9, lambda_: float = 0.8) -> torch.Tensor:
    """0.5 * MSE(lambda-return, V[:-1]); value [T+1,B], reward [T,B]."""
    value, reward, weight = data
    if weight is None:
        weight = torch.ones_like(reward)
    with torch.no_grad():
        return_ = generalized_lambda_returns(value, reward, gamma, lambda_)
    return 0.5 * (F.mse_loss(return_, value[:-1], reduction='none') * weight).mean()


# --------------------------------------------------------------- q-learning
def q_1step_td_error(
    data: namedtuple,
    gamma: float,
    criterion=nn.MSELoss(reduction='none'),
) -> torch.Tensor:
    q, next_q, act, next_act, reward, done, weight = data
    if weight is None:
        weight = torch.ones_like(reward)
    batch_range = torch.arange(act.shape[0])
    q_s_a = q[batch_range, act]
    target_q_s_a = next_q[batch_range, next_act]
    target = reward + gamma * target_q_s_a * (1 - done.float())
    return (criterion(q_s_a, target.detach()) * weight).mean()


def m_q_1step_td_error(
    data: namedtuple,
    gamma: float,
    tau: float,
    alpha: float,
    criterion=nn.MSELoss(reduction='none'),
) -> torch.Tensor:
    """Munchausen DQN 1-step TD (log-policy augmented reward).

    Parity: reference td.py:78 (policy/mdqn.py).
    """
    q, target_q, next_q, act, reward, done, weight = data
    lower = -1.0
    if weight is None:
        weight = torch.ones_like(reward)
    batch_range = torch.arange(act.shape[0])
    q_s_a = q[batch_range, act]

    # target policy log-prob of taken action (softmax with temperature tau)
    target_v_s = target_q.max(dim=-1, keepdim=True)[0]
    logsum = torch.logsumexp((target_q - target_v_s) / tau, dim=-1, keepdim=True)
    log_pi_a = target_q - target_v_s - tau * logsum  # [B, N]
    munchausen = alpha * torch.clamp(log_pi_a[batch_range, act], min=lower, max=1)

    # soft TD target from next state
    next_v_s = next_q.max(dim=-1, keepdim=True)[0]
    next_logsum = torch.logsumexp((next_q - next_v_s) / tau, dim=-1, keepdim=True)
    next_log_pi = next_q - next_v_s - tau * next_logsum
    next_pi = torch.softmax((next_q - next_v_s) / tau, dim=-1)
    soft_next = (next_pi * (next_q - next_log_pi)).sum(dim=-1)

    target = reward + munchausen + gamma * soft_next * (1 - done.float())
    td_error_per_sample = criterion(q_s_a, target.detach())
    return (td_error_per_sample * weight).mean(), td_error_per_sample, q_s_a.detach().mean()


def q_v_1step_td_error(data: namedtuple, gamma: float, criterion=nn.MSELoss(reduction='none')) -> torch.Tensor:
    """Q towards r + gamma V(s') (discrete SAC critic). Parity: td.py:164."""
    q, v, act, reward, done, weight = data
    if weight is None:
        weight = torch.ones_like(reward)
    if len(act.shape) == 1:
        batch_range = torch.arange(act.shape[0])
        q_s_a = q[batch_range, act]
    else:  # MARL
        q_s_a = q.gather(-1, act.unsqueeze(-1)).squeeze(-1)
        reward = reward.unsqueeze(-1)
        done = done.unsqueeze(-1)
        weight = weight.unsqueeze(-1)
    target = reward + gamma * v * (1 - done.float())
    td_error_per_sample = criterion(q_s_a, target.detach())
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def q_nstep_td_error(
    data: namedtuple,
    gamma: Union[float, list],
    nstep: int = 1,
    cum_reward: bool = False,
    value_gamma: Optional[torch.Tensor] = None,
    criterion=nn.MSELoss(reduction='none'),
):
    """n-step TD for Q-learning. q/next_n_q [B,N] (or [B,A,N] MARL), action
    [B] (or [B,A]), reward [T,B], done [B]. Returns (loss, per-sample-td).
    """
    q, next_n_q, action, next_n_action, reward, done, weight = data
    if weight is None:
        weight = torch.ones_like(reward[0] if reward.dim() > 1 else reward)
    marl = action.dim() > 1
    # fused HIP lane: single-launch forward {td, return, q_sa}; one-scatter
    # backward (see ding/ops/csrc/td_ops.hip)
    from ding.ops import dispatch as _dispatch
    if (
        not marl and not cum_reward and isinstance(criterion, nn.MSELoss) and isinstance(gamma, float)
        and isinstance(q, torch.Tensor) and q.dim() == 2 and q.dtype == torch.float32
        and reward.dim() == 2 and _dispatch.use_hip_autograd(q)
    ):
        td, _ret = _dispatch.fused_q_nstep_td(
            q, next_n_q, action, next_n_action, reward, done, value_gamma, gamma, nstep, rescale=False
        )
        td_error_per_sample = td.pow(2)
        return (td_error_per_sample * weight).mean(), td_error_per_sample
    if not marl:
        action_ = action.unsqueeze(-1)
    else:
        action_ = action.unsqueeze(-1)
        reward = reward.unsqueeze(-1)
        weight = weight.unsqueeze(-1)
        done = done.unsqueeze(-1)
        if value_gamma is not None:
            value_gamma = value_gamma.unsqueeze(-1)
    q_s_a = q.gather(-1, action_).squeeze(-1)
    target_q_s_a = next_n_q.gather(-1, next_n_action.unsqueeze(-1)).squeeze(-1)
    if cum_reward:
        if value_gamma is None:
            target_q_s_a = reward + (gamma ** nstep) * target_q_s_a * (1 - done.float())
        else:
            target_q_s_a = reward + value_gamma * target_q_s_a * (1 - done.float())
    else:
        target_q_s_a = nstep_return(nstep_return_data(reward, target_q_s_a, done), gamma, nstep, value_gamma)
    td_error_per_sample = criterion(q_s_a, target_q_s_a.detach())
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def q_nstep_td_error_with_rescale(
    data: namedtuple,
    gamma: Union[float, list],
    nstep: int = 1,
    value_gamma: Optional[torch.Tensor] = None,
    criterion=nn.MSELoss(reduction='none'),
    trans_fn=value_transform,
    inv_trans_fn=value_inv_transform,
):
    """n-step TD with R2D2 value rescale: target in h-space,
    h(G + gamma^n h^-1(Q'))."""
    q, next_n_q, action, next_n_action, reward, done, weight = data
    if weight is None:
        weight = torch.ones_like(action, dtype=q.dtype)
    from ding.ops import dispatch as _dispatch
    if (
        isinstance(criterion, nn.MSELoss) and isinstance(gamma, float) and trans_fn is value_transform
        and inv_trans_fn is value_inv_transform and isinstance(q, torch.Tensor) and q.dim() == 2
        and q.dtype == torch.float32 and reward.dim() == 2 and _dispatch.use_hip_autograd(q)
    ):
        td, _ret = _dispatch.fused_q_nstep_td(
            q, next_n_q, action, next_n_action, reward, done, value_gamma, gamma, nstep, rescale=True
        )
        td_error_per_sample = td.pow(2)
        return (td_error_per_sample * weight).mean(), td_error_per_sample
    q_s_a = q.gather(-1, action.unsqueeze(-1)).squeeze(-1)
    target_q_s_a = next_n_q.gather(-1, next_n_action.unsqueeze(-1)).squeeze(-1)
    target_q_s_a = inv_trans_fn(target_q_s_a)
    target_q_s_a = nstep_return(nstep_return_data(reward, target_q_s_a, done), gamma, nstep, value_gamma)
    target_q_s_a = trans_fn(target_q_s_a)
    td_error_per_sample = criterion(q_s_a, target_q_s_a.detach())
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def bdq_nstep_td_error(
    data: namedtuple,
    gamma: Union[float, list],
    nstep: int = 1,
    cum_reward: bool = False,
    value_gamma: Optional[torch.Tensor] = None,
    criterion=nn.MSELoss(reduction='none'),
):
    """Branching-DQN n-step TD: q [B, D, m] with D branches; loss averaged
    over branches. Parity: td.py:722."""
    q, next_n_q, action, next_n_action, reward, done, weight = data
    if weight is None:
        weight = torch.ones_like(reward)
    reward = reward.unsqueeze(-1)
    done = done.unsqueeze(-1)
    if value_gamma is not None:
        value_gamma = value_gamma.unsqueeze(-1)
    q_s_a = q.gather(-1, action.unsqueeze(-1)).squeeze(-1)  # [B, D]
    target_q_s_a = next_n_q.gather(-1, next_n_action.unsqueeze(-1)).squeeze(-1)
    if cum_reward:
        g = value_gamma if value_gamma is not None else gamma ** nstep
        target_q_s_a = reward + g * target_q_s_a * (1 - done.float())
    else:
        target_q_s_a = nstep_return(nstep_return_data(reward, target_q_s_a, done), gamma, nstep, value_gamma)
    td_error_per_sample = criterion(q_s_a, target_q_s_a.detach()).mean(-1)
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def dqfd_nstep_td_error(
    data: namedtuple,
    gamma: float,
    lambda_n_step_td: float,
    lambda_supervised_loss: float,
    margin_function: float,
    lambda_one_step_td: float = 1.0,
    nstep: int = 1,
    cum_reward: bool = False,
    value_gamma: Optional[torch.Tensor] = None,
    criterion=nn.MSELoss(reduction='none'),
):
    """DQfD loss: 1-step TD + lambda_n * n-step TD + lambda_E * large-margin
    supervised loss on expert transitions. Parity: td.py:870."""
    q, next_n_q, action, next_n_action, reward, done, done_one_step, weight, next_q_one_step, \
        next_action_one_step, is_expert = data
    if weight is None:
        weight = torch.ones_like(action, dtype=q.dtype)
    batch_range = torch.arange(action.shape[0])
    q_s_a = q[batch_range, action]
    # n-step
    target_q_s_a = next_n_q[batch_range, next_n_action]
    target_q_s_a = nstep_return(nstep_return_data(reward, target_q_s_a, done), gamma, nstep, value_gamma)
    td_n = criterion(q_s_a, target_q_s_a.detach())
    # one-step
    target_q_one = next_q_one_step[batch_range, next_action_one_step]
    target_q_one = reward[0] + gamma * target_q_one * (1 - done_one_step.float())
    td_1 = criterion(q_s_a, target_q_one.detach())
    # large-margin supervised loss
    n_action = q.shape[-1]
    margin = torch.full_like(q, margin_function)
    margin[batch_range, action] = 0.0
    l_margin = (q + margin).max(dim=-1)[0] - q_s_a
    sl = is_expert.float() * l_margin
    loss_per_sample = lambda_one_step_td * td_1 + lambda_n_step_td * td_n + lambda_supervised_loss * sl
    return (loss_per_sample * weight).mean(), loss_per_sample, (td_1.mean(), td_n.mean(), sl.mean())


def dqfd_nstep_td_error_with_rescale(
    data: namedtuple,
    gamma: float,
    lambda_n_step_td: float,
    lambda_supervised_loss: float,
    lambda_one_step_td: float,
    margin_function: float,
    nstep: int = 1,
    cum_reward: bool = False,
    value_gamma: Optional[torch.Tensor] = None,
    criterion=nn.MSELoss(reduction='none'),
    trans_fn=value_transform,
    inv_trans_fn=value_inv_transform,
):
    q, next_n_q, action, next_n_action, reward, done, done_one_step, weight, next_q_one_step, \
        next_action_one_step, is_expert = data
    if weight is None:
        weight = torch.ones_like(action, dtype=q.dtype)
    batch_range = torch.arange(action.shape[0])
    q_s_a = q[batch_range, action]
    target_q_s_a = inv_trans_fn(next_n_q[batch_range, next_n_action])
    target_q_s_a = trans_fn(nstep_return(nstep_return_data(reward, target_q_s_a, done), gamma, nstep, value_gamma))
    td_n = criterion(q_s_a, target_q_s_a.detach())
    target_q_one = inv_trans_fn(next_q_one_step[batch_range, next_action_one_step])
    target_q_one = trans_fn(reward[0] + gamma * target_q_one * (1 - done_one_step.float()))
    td_1 = criterion(q_s_a, target_q_one.detach())
    margin = torch.full_like(q, margin_function)
    margin[batch_range, action] = 0.0
    sl = is_expert.float() * ((q + margin).max(dim=-1)[0] - q_s_a)
    loss_per_sample = lambda_one_step_td * td_1 + lambda_n_step_td * td_n + lambda_supervised_loss * sl
    return (loss_per_sample * weight).mean(), loss_per_sample, (td_1.mean(), td_n.mean(), sl.mean())


def q_nstep_sql_td_error(
    data: namedtuple,
    gamma: float,
    alpha: float,
    nstep: int = 1,
    cum_reward: bool = False,
    value_gamma: Optional[torch.Tensor] = None,
    criterion=nn.MSELoss(reduction='none'),
):
    """Soft Q-learning n-step TD: target uses alpha*logsumexp(Q'/alpha).
    Parity: td.py:1175."""
    q, next_n_q, action, next_n_action, reward, done, weight = data
    if weight is None:
        weight = torch.ones_like(action, dtype=q.dtype)
    batch_range = torch.arange(action.shape[0])
    q_s_a = q[batch_range, action]
    target_v = alpha * torch.logsumexp(next_n_q / alpha, dim=-1)
    if cum_reward:
        g = value_gamma if value_gamma is not None else gamma ** nstep
        target = reward + g * target_v * (1 - done.float())
    else:
        target = nstep_return(nstep_return_data(reward, target_v, done), gamma, nstep, value_gamma)
    td_error_per_sample = criterion(q_s_a, target.detach())
    return (td_error_per_sample * weight).mean(), td_error_per_sample


# ------------------------------------------------------------ distributional
def dist_1step_td_error(data: namedtuple, gamma: float, v_min: float, v_max: float, n_atom: int) -> torch.Tensor:
    dist, next_dist, act, next_act, reward, done, weight = data
    reward = reward.unsqueeze(-1)
    done = done.unsqueeze(-1)
    return _categorical_projection_loss(
        dist, next_dist, act, next_act, reward, done, weight, v_min, v_max, n_atom, gamma
    )[0]


def _categorical_projection_loss(dist, next_dist, act, next_act, reward, done, weight, v_min, v_max, n_atom, gamma_n):
    """Project r + gamma_n * z onto the fixed support; cross-entropy loss.

    dist/next_dist: [B, N, n_atom]; act/next_act: [B]; reward/done: [B, 1].
    gamma_n may be a scalar (gamma**nstep) or a [B,1] tensor.
    """
    device = reward.device
    batch_size = act.shape[0]
    batch_range = torch.arange(batch_size, device=device)
    support = torch.linspace(v_min, v_max, n_atom, device=device)
    delta_z = (v_max - v_min) / (n_atom - 1)
    next_p = next_dist[batch_range, next_act].detach()  # [B, n_atom]
    target_z = reward + (1 - done.float()) * gamma_n * support  # [B, n_atom]
    target_z = target_z.clamp(min=v_min, max=v_max)
    b = (target_z - v_min) / delta_z
    l = b.floor().long()
    u = b.ceil().long()
    l[(u > 0) & (l == u)] -= 1
    u[(l < (n_atom - 1)) & (l == u)] += 1
    proj = torch.zeros_like(next_p)
    offset = (torch.arange(batch_size, device=device) * n_atom).unsqueeze(1).expand(batch_size, n_atom)
    proj.view(-1).index_add_(0, (l + offset).reshape(-1), (next_p * (u.float() - b)).reshape(-1))
    proj.view(-1).index_add_(0, (u + offset).reshape(-1), (next_p * (b - l.float())).reshape(-1))
    log_p = torch.log(dist[batch_range, act].clamp(min=1e-20))
    td_error_per_sample = -(log_p * proj).sum(-1)
    if weight is None:
        w = torch.ones_like(td_error_per_sample)
    else:
        w = weight if isinstance(weight, torch.Tensor) else torch.tensor(weight, device=device)
        if w.dim() > 1:
            w = w.squeeze(-1)
    loss = (td_error_per_sample * w).mean()
    return loss, td_error_per_sample


def dist_nstep_td_error(
    data: namedtuple,
    gamma: float,
    v_min: float,
    v_max: float,
    n_atom: int,
    nstep: int = 1,
    value_gamma: Optional[torch.Tensor] = None,
):
    """C51 n-step TD (categorical projection + cross-entropy).

    dist [B,N,n_atom], reward [T,B]. Returns (loss, td_error_per_sample).
    """
    from ding.ops import dispatch
    dist, next_n_dist, act, next_n_act, reward, done, weight = data
    device = reward.device
    factor = gamma ** torch.arange(nstep, dtype=reward.dtype, device=device)
    reward_n = torch.matmul(factor, reward)  # [B]
    if dispatch.use_hip(next_n_dist.detach()) and act.dim() == 1 and value_gamma is None:
        # HIP kernel computes the categorical projection (grad-free); the
        # differentiable cross-entropy stays in torch.
        proj = dispatch.c51_project(next_n_dist.detach(), next_n_act, reward_n, done, v_min, v_max, gamma ** nstep)
        batch_range = torch.arange(act.shape[0], device=device)
        log_p = torch.log(dist[batch_range, act].clamp(min=1e-20))
        td_error_per_sample = -(log_p * proj).sum(-1)
        w = torch.ones_like(td_error_per_sample) if weight is None else weight
        return (td_error_per_sample * w).mean(), td_error_per_sample
    if act.dim() == 1:
        reward_b = reward_n.unsqueeze(-1)
        done_b = done.unsqueeze(-1)
        if value_gamma is None:
            g = gamma ** nstep
        else:
            g = value_gamma.unsqueeze(-1) if isinstance(value_gamma, torch.Tensor) else value_gamma
        return _categorical_projection_loss(
            dist, next_n_dist, act, next_n_act, reward_b, done_b, weight, v_min, v_max, n_atom, g
        )
    else:  # MARL: flatten [B, A] agents into batch
        B, A = act.shape
        N = dist.shape[2]
        dist_f = dist.reshape(B * A, N, -1)
        next_f = next_n_dist.reshape(B * A, N, -1)
        act_f = act.reshape(B * A)
        next_act_f = next_n_act.reshape(B * A)
        reward_b = reward_n.unsqueeze(-1).repeat(1, A).reshape(B * A, 1)
        done_b = done.unsqueeze(-1).repeat(1, A).reshape(B * A, 1)
        g = gamma ** nstep if value_gamma is None else value_gamma.unsqueeze(-1).repeat(1, A).reshape(B * A, 1)
        w = None if weight is None else weight.unsqueeze(-1).repeat(1, A).reshape(B * A)
        return _categorical_projection_loss(
            dist_f, next_f, act_f, next_act_f, reward_b, done_b, w, v_min, v_max, n_atom, g
        )


# ---------------------------------------------------------------- v-learning
def v_1step_td_error(data: namedtuple, gamma: float, criterion=nn.MSELoss(reduction='none')):
    v, next_v, reward, done, weight = data
    if weight is None:
        weight = torch.ones_like(v)
    if done is not None:
        target_v = reward + gamma * next_v * (1 - done.float())
    else:
        target_v = reward + gamma * next_v
    td_error_per_sample = criterion(v, target_v.detach())
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def v_nstep_td_error(data: namedtuple, gamma: float, nstep: int = 1, criterion=nn.MSELoss(reduction='none')):
    """value [B], next_n_v [B], reward [T,B]. Parity: td.py:579."""
    v, next_n_v, reward, done, weight, value_gamma = data
    if weight is None:
        weight = torch.ones_like(v)
    target_v = nstep_return(nstep_return_data(reward, next_n_v, done), gamma, nstep, value_gamma)
    td_error_per_sample = criterion(v, target_v.detach())
    return (td_error_per_sample * weight).mean(), td_error_per_sample


# ------------------------------------------------------------------ quantile
def evaluate_quantile_at_action(q_s, actions):
    """q_s: [B, num_quantiles, N] -> [B, num_quantiles] at the taken action."""
    B, num_q = q_s.shape[:2]
    a = actions.view(B, 1, 1).expand(B, num_q, 1)
    return q_s.gather(-1, a).squeeze(-1)


def _quantile_huber_loss(pred, target, taus, kappa: float = 1.0):
    """pred [B, Nq], target [B, Nq'], taus [B, Nq] -> per-sample loss [B]."""
    diff = target.unsqueeze(1) - pred.unsqueeze(2)  # [B, Nq, Nq']
    if kappa > 0:
        huber = torch.where(diff.abs() <= kappa, 0.5 * diff.pow(2), kappa * (diff.abs() - 0.5 * kappa))
    else:
        huber = diff.abs()
    rho = (taus.unsqueeze(2) - (diff.detach() < 0).float()).abs() * huber
    if kappa > 0:
        rho = rho / kappa
    return rho.sum(1).mean(1)


def qrdqn_nstep_td_error(
    data: namedtuple,
    gamma: float,
    nstep: int = 1,
    value_gamma: Optional[torch.Tensor] = None,
):
    """QR-DQN n-step TD. q: [B, tau, N]. Parity: td.py:1098."""
    q, next_n_q, action, next_n_action, reward, done, tau, weight = data
    if weight is None:
        weight = torch.ones_like(action, dtype=q.dtype)
    B, num_q = q.shape[0], q.shape[1]
    q_s_a = evaluate_quantile_at_action(q, action)  # [B, tau]
    target_q_s_a = evaluate_quantile_at_action(next_n_q, next_n_action)
    target = nstep_return(
        nstep_return_data(reward.unsqueeze(-1), target_q_s_a, done.unsqueeze(-1)), gamma, nstep,
        value_gamma if value_gamma is None else value_gamma.unsqueeze(-1)
    ).detach()
    taus = (torch.arange(num_q, device=q.device, dtype=q.dtype) + 0.5) / num_q
    taus = taus.unsqueeze(0).expand(B, num_q)
    td_error_per_sample = _quantile_huber_loss(q_s_a, target, taus)
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def iqn_nstep_td_error(
    data: namedtuple,
    gamma: float,
    nstep: int = 1,
    kappa: float = 1.0,
    value_gamma: Optional[torch.Tensor] = None,
):
    """IQN n-step TD. q: [tau, B, N]; replay_quantiles: [tau, B, 1].
    Parity: td.py:1253."""
    q, next_n_q, action, next_n_action, reward, done, replay_quantiles, weight = data
    if weight is None:
        weight = torch.ones_like(action, dtype=q.dtype)
    # reorder [tau, B, N] -> [B, tau, N]
    q_b = q.permute(1, 0, 2)
    next_b = next_n_q.permute(1, 0, 2)
    B, num_q = q_b.shape[0], q_b.shape[1]
    q_s_a = evaluate_quantile_at_action(q_b, action)
    target_q_s_a = evaluate_quantile_at_action(next_b, next_n_action)
    target = nstep_return(
        nstep_return_data(reward.unsqueeze(-1), target_q_s_a, done.unsqueeze(-1)), gamma, nstep,
        value_gamma if value_gamma is None else value_gamma.unsqueeze(-1)
    ).detach()
    taus = replay_quantiles.squeeze(-1).t()  # [B, tau]
    td_error_per_sample = _quantile_huber_loss(q_s_a, target, taus, kappa)
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def fqf_nstep_td_error(
    data: namedtuple,
    gamma: float,
    nstep: int = 1,
    kappa: float = 1.0,
    value_gamma: Optional[torch.Tensor] = None,
):
    """FQF n-step TD. q: [B, tau, N]; quantiles_hats: [B, tau].
    Parity: td.py:1359."""
    q, next_n_q, action, next_n_action, reward, done, quantiles_hats, weight = data
    if weight is None:
        weight = torch.ones_like(action, dtype=q.dtype)
    q_s_a = evaluate_quantile_at_action(q, action)
    target_q_s_a = evaluate_quantile_at_action(next_n_q, next_n_action)
    target = nstep_return(
        nstep_return_data(reward.unsqueeze(-1), target_q_s_a, done.unsqueeze(-1)), gamma, nstep,
        value_gamma if value_gamma is None else value_gamma.unsqueeze(-1)
    ).detach()
    td_error_per_sample = _quantile_huber_loss(q_s_a, target, quantiles_hats, kappa)
    return (td_error_per_sample * weight).mean(), td_error_per_sample


def fqf_calculate_fraction_loss(q_tau_i, q_value, quantiles, actions):
    """FQF fraction-proposal gradient loss. q_tau_i: [B, tau-1, N] at interior
    quantiles; q_value: [B, tau, N]; quantiles: [B, tau+1]."""
    B = q_value.shape[0]
    num_q = q_value.shape[1]
    batch_range = torch.arange(B, device=q_value.device)
    sa_tau_i = q_tau_i[batch_range, :, actions]  # [B, tau-1]
    sa_q = q_value[batch_range, :, actions]  # [B, tau]
    # gradient of W1 distance wrt interior quantile fractions
    values_1 = sa_tau_i - sa_q[:, :-1]
    signs_1 = sa_tau_i > torch.cat([sa_q[:, :1], sa_tau_i[:, :-1]], dim=1)
    values_2 = sa_tau_i - sa_q[:, 1:]
    signs_2 = sa_tau_i < torch.cat([sa_tau_i[:, 1:], sa_q[:, -1:]], dim=1)
    grad = torch.where(signs_1, values_1, -values_1) + torch.where(signs_2, values_2, -values_2)
    grad = grad.view(B, num_q - 1).detach()
    return (grad * quantiles[:, 1:-1]).sum(dim=1).mean()
