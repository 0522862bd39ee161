"""HAPPO losses (heterogeneous-agent PPO with sequential-update factor).

Parity: reference ding/rl_utils/happo.py:18,81,150,195. Identical to PPO
except the surrogate is multiplied by the running ``factor`` from previously
updated agents.
"""
from collections import namedtuple
from typing import Optional, Tuple

import torch

from .ppo import ppo_value_data, ppo_value_error, ppo_info, _gaussian_dist

happo_value_data = namedtuple('happo_value_data', ['value_new', 'value_old', 'return_', 'weight'])
happo_loss = namedtuple('happo_loss', ['policy_loss', 'value_loss', 'entropy_loss'])
happo_policy_loss = namedtuple('happo_policy_loss', ['policy_loss', 'entropy_loss'])
happo_info = namedtuple('happo_info', ['approx_kl', 'clipfrac'])
happo_data = namedtuple(
    'happo_data', ['logit_new', 'logit_old', 'action', 'value_new', 'value_old', 'adv', 'return_', 'weight', 'factor']
)
happo_policy_data = namedtuple('happo_policy_data', ['logit_new', 'logit_old', 'action', 'adv', 'weight', 'factor'])


def happo_policy_error(
    data: namedtuple,
    clip_ratio: float = 0.2,
    dual_clip: Optional[float] = None,
) -> Tuple[namedtuple, namedtuple]:
    logit_new, logit_old, action, adv, weight, factor = data
    if weight is None:
        weight = torch.ones_like(adv)
    dist_new = torch.distributions.Categorical(logits=logit_new)
    dist_old = torch.distributions.Categorical(logits=logit_old)
    logp_new = dist_new.log_prob(action)
    logp_old = dist_old.log_prob(action)
    entropy_loss = (dist_new.entropy() * weight).mean()
    ratio = torch.exp(logp_new - logp_old)
    surr1 = ratio * adv
    surr2 = ratio.clamp(1 - clip_ratio, 1 + clip_ratio) * adv
    if dual_clip is not None:
        assert dual_clip > 1.0
        clipped = torch.min(surr1, surr2)
        body = torch.where(adv < 0, torch.max(clipped, dual_clip * adv), clipped)
    else:
        body = torch.min(surr1, surr2)
    policy_loss = -(factor.squeeze(-1) * body * weight).mean()
    with torch.no_grad():
        approx_kl = (logp_old - logp_new).mean().item()
        clipfrac = ((ratio - 1.0).abs() > clip_ratio).float().mean().item()
    return happo_policy_loss(policy_loss, entropy_loss), happo_info(approx_kl, clipfrac)


def happo_value_error(data: namedtuple, clip_ratio: float = 0.2, use_value_clip: bool = True) -> torch.Tensor:
    return ppo_value_error(ppo_value_data(*data), clip_ratio, use_value_clip)


def happo_error(
    data: namedtuple,
    clip_ratio: float = 0.2,
    use_value_clip: bool = True,
    dual_clip: Optional[float] = None,
) -> Tuple[namedtuple, namedtuple]:
    logit_new, logit_old, action, value_new, value_old, adv, return_, weight, factor = data
    pol, info = happo_policy_error(
        happo_policy_data(logit_new, logit_old, action, adv, weight, factor), clip_ratio, dual_clip
    )
    v_loss = happo_value_error(happo_value_data(value_new, value_old, return_, weight), clip_ratio, use_value_clip)
    return happo_loss(pol.policy_loss, v_loss, pol.entropy_loss), info


def happo_policy_error_continuous(
    data: namedtuple,
    clip_ratio: float = 0.2,
    dual_clip: Optional[float] = None,
) -> Tuple[namedtuple, namedtuple]:
    """Policy-only continuous HAPPO surrogate (factor-weighted).
    Parity: reference ding/rl_utils/happo.py happo_policy_error_continuous:287."""
    mu_sigma_new, mu_sigma_old, action, adv, weight, factor = data
    if weight is None:
        weight = torch.ones_like(adv)
    dist_new = _gaussian_dist(mu_sigma_new)
    dist_old = _gaussian_dist(mu_sigma_old)
    logp_new = dist_new.log_prob(action)
    logp_old = dist_old.log_prob(action)
    entropy_loss = (dist_new.entropy() * weight).mean()
    ratio = torch.exp(logp_new - logp_old)
    surr1 = ratio * adv
    surr2 = ratio.clamp(1 - clip_ratio, 1 + clip_ratio) * adv
    if dual_clip is not None:
        clipped = torch.min(surr1, surr2)
        body = torch.where(adv < 0, torch.max(clipped, dual_clip * adv), clipped)
    else:
        body = torch.min(surr1, surr2)
    policy_loss = -(factor.squeeze(-1) * body * weight).mean()
    with torch.no_grad():
        approx_kl = (logp_old - logp_new).mean().item()
        clipfrac = ((ratio - 1.0).abs() > clip_ratio).float().mean().item()
    return happo_policy_loss(policy_loss, entropy_loss), happo_info(approx_kl, clipfrac)


def happo_error_continuous(
    data: namedtuple,
    clip_ratio: float = 0.2,
    use_value_clip: bool = True,
    dual_clip: Optional[float] = None,
) -> Tuple[namedtuple, namedtuple]:
    mu_sigma_new, mu_sigma_old, action, value_new, value_old, adv, return_, weight, factor = data
    pol, info = happo_policy_error_continuous(
        happo_policy_data(mu_sigma_new, mu_sigma_old, action, adv, weight, factor), clip_ratio, dual_clip
    )
    v_loss = ppo_value_error(ppo_value_data(value_new, value_old, return_, weight), clip_ratio, use_value_clip)
    return happo_loss(pol.policy_loss, v_loss, pol.entropy_loss), info
