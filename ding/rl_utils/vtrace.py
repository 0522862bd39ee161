"""V-trace (IMPALA) losses, discrete and continuous.

Parity: reference ding/rl_utils/vtrace.py (vtrace_nstep_return:9,
vtrace_advantage:32, vtrace_error_discrete_action:73). On GPU the clipped
reverse scan dispatches to the HIP kernel in ding/ops/csrc/scan_ops.hip via
ding/ops/dispatch.py; IS ratios and the three-term loss stay in PyTorch here —
the fully fused path lives in vtrace_error_discrete_fused (vtrace_ops.hip)
when available.
"""
from collections import namedtuple

import torch
import torch.nn.functional as F
from torch.distributions import Categorical, Independent, Normal

from .isw import compute_importance_weights

vtrace_data = namedtuple('vtrace_data', ['target_output', 'behaviour_output', 'action', 'value', 'reward', 'weight'])
vtrace_loss = namedtuple('vtrace_loss', ['policy_loss', 'value_loss', 'entropy_loss'])


def vtrace_nstep_return(clipped_rhos, clipped_cs, reward, bootstrap_values, gamma=0.99, lambda_=0.95):
    """vs: reverse scan of delta_t = rho_t (r_t + gamma V_{t+1} - V_t),
    vs_t = V_t + delta_t + gamma c_t (vs_{t+1} - V_{t+1}).

    bootstrap_values: [T+1, B]; everything else [T, B].
    """
    from ding.ops import dispatch
    deltas = clipped_rhos * (reward + gamma * bootstrap_values[1:] - bootstrap_values[:-1])
    factor = gamma * lambda_
    if dispatch.use_hip(deltas):
        corr = dispatch.gae_scan(deltas, factor * clipped_cs)
        return bootstrap_values[:-1] + corr
    result = bootstrap_values[:-1].clone()
    acc = torch.zeros_like(result[0])
    for t in range(reward.shape[0] - 1, -1, -1):
        acc = deltas[t] + factor * clipped_cs[t] * acc
        result[t] = result[t] + acc
    return result


def vtrace_advantage(clipped_pg_rhos, reward, return_, bootstrap_values, gamma):
    """pg advantage: rho^pg_t (r_t + gamma vs_{t+1} - V_t)."""
    return clipped_pg_rhos * (reward + gamma * return_ - bootstrap_values)


def vtrace_error_discrete_action(
    data: namedtuple,
    gamma: float = 0.99,
    lambda_: float = 0.95,
    rho_clip_ratio: float = 1.0,
    c_clip_ratio: float = 1.0,
    rho_pg_clip_ratio: float = 1.0,
):
    """IMPALA loss. target/behaviour_output: logits [T,B,N]; action [T,B];
    value [T+1,B]; reward [T,B]. Returns (pg_loss, value_loss, entropy_loss).
    """
    target_output, behaviour_output, action, value, reward, weight = data
    from ding.ops import dispatch
    if dispatch.use_hip_autograd(target_output):
        pg_loss, value_loss, entropy_loss = dispatch.fused_vtrace_error(
            target_output.float(), behaviour_output.float(), action, value.float(), reward.float(),
            None if weight is None else weight.float(), gamma, lambda_, rho_clip_ratio, c_clip_ratio,
            rho_pg_clip_ratio
        )
        return vtrace_loss(pg_loss, value_loss, entropy_loss)
    with torch.no_grad():
        IS = compute_importance_weights(target_output, behaviour_output, action, 'discrete')
        rhos = IS.clamp(max=rho_clip_ratio)
        cs = IS.clamp(max=c_clip_ratio)
        return_ = vtrace_nstep_return(rhos, cs, reward, value, gamma, lambda_)
        pg_rhos = IS.clamp(max=rho_pg_clip_ratio)
        return_t_plus_1 = torch.cat([return_[1:], value[-1:]], 0)
        adv = vtrace_advantage(pg_rhos, reward, return_t_plus_1, value[:-1], gamma)
    if weight is None:
        weight = torch.ones_like(reward)
    logp = F.log_softmax(target_output, -1)
    logp_a = logp.gather(-1, action.long().unsqueeze(-1)).squeeze(-1)
    pg_loss = -(logp_a * adv * weight).mean()
    value_loss = (F.mse_loss(value[:-1], return_, reduction='none') * weight).mean()
    entropy = -(logp.exp() * logp).sum(-1)
    entropy_loss = (entropy * weight).mean()
    return vtrace_loss(pg_loss, value_loss, entropy_loss)


def vtrace_error_continuous_action(
    data: namedtuple,
    gamma: float = 0.99,
    lambda_: float = 0.95,
    rho_clip_ratio: float = 1.0,
    c_clip_ratio: float = 1.0,
    rho_pg_clip_ratio: float = 1.0,
):
    """Continuous-action v-trace; outputs are {'mu','sigma'} dicts [T,B,D]."""
    target_output, behaviour_output, action, value, reward, weight = data
    with torch.no_grad():
        IS = compute_importance_weights(target_output, behaviour_output, action, 'continuous')
        rhos = IS.clamp(max=rho_clip_ratio)
        cs = IS.clamp(max=c_clip_ratio)
        return_ = vtrace_nstep_return(rhos, cs, reward, value, gamma, lambda_)
        pg_rhos = IS.clamp(max=rho_pg_clip_ratio)
        return_t_plus_1 = torch.cat([return_[1:], value[-1:]], 0)
        adv = vtrace_advantage(pg_rhos, reward, return_t_plus_1, value[:-1], gamma)
    if weight is None:
        weight = torch.ones_like(reward)
    dist_target = Independent(Normal(target_output['mu'], target_output['sigma']), 1)
    pg_loss = -(dist_target.log_prob(action) * adv * weight).mean()
    value_loss = (F.mse_loss(value[:-1], return_, reduction='none') * weight).mean()
    entropy_loss = (dist_target.entropy() * weight).mean()
    return vtrace_loss(pg_loss, value_loss, entropy_loss)
