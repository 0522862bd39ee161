"""PPO losses (discrete + continuous), with value clip, dual clip and the
RLHF KL estimators.

Parity: reference ding/rl_utils/ppo.py (ppo_error:77, ppo_policy_error:143,
ppo_value_error:233, ppo_error_continuous:278). GPU [B,N] logits dispatch to
the fused HIP kernel (ding/ops/csrc/ppo_ops.hip) which computes log-softmax,
ratio, clipped surrogates, entropy and value clip in one pass over HBM.
"""
from collections import namedtuple
from typing import Optional, Tuple

import torch

ppo_data = namedtuple(
    'ppo_data',
    ['logit_new', 'logit_old', 'action', 'value_new', 'value_old', 'adv', 'return_', 'weight', 'logit_pretrained']
)
# keep old 8-field construction working (logit_pretrained optional)
ppo_data.__new__.__defaults__ = (None, )
ppo_data_continuous = namedtuple(
    'ppo_data_continuous',
    ['mu_sigma_new', 'mu_sigma_old', 'action', 'value_new', 'value_old', 'adv', 'return_', 'weight']
)
ppo_policy_data = namedtuple('ppo_policy_data', ['logit_new', 'logit_old', 'action', 'adv', 'weight', 'logit_pretrained'])
ppo_policy_data.__new__.__defaults__ = (None, )
ppo_policy_data_continuous = namedtuple(
    'ppo_policy_data_continuous', ['mu_sigma_new', 'mu_sigma_old', 'action', 'adv', 'weight']
)
ppo_value_data = namedtuple('ppo_value_data', ['value_new', 'value_old', 'return_', 'weight'])
ppo_loss = namedtuple('ppo_loss', ['policy_loss', 'value_loss', 'entropy_loss', 'kl_div'])
ppo_policy_loss = namedtuple('ppo_policy_loss', ['policy_loss', 'entropy_loss', 'kl_div'])
ppo_info = namedtuple('ppo_info', ['approx_kl', 'clipfrac'])


def calculate_kl_div(log_ratio: torch.Tensor, kl_type: str) -> torch.Tensor:
    """Schulman KL estimators; log_ratio = logp_new - logp_pretrained."""
    if kl_type == 'k1':
        return log_ratio.mean()
    if kl_type == 'k2':
        return (log_ratio ** 2 / 2).mean()
    if kl_type == 'k3':
        return (torch.exp(-log_ratio) - 1 + log_ratio).mean()
    raise ValueError(f"unknown kl_type: {kl_type}")


def ppo_policy_error(
    data: namedtuple,
    clip_ratio: float = 0.2,
    dual_clip: Optional[float] = None,
    entropy_bonus: bool = True,
    kl_type: str = 'k1',
) -> Tuple[namedtuple, namedtuple]:
    """Clipped-surrogate policy loss + entropy (+optional pretrained-KL)."""
    logit_new, logit_old, action, adv, weight = data[:5]
    logit_pretrained = data[5] if len(data) > 5 else None
    if weight is None:
        weight = torch.ones_like(adv)
    dist_new = torch.distributions.Categorical(logits=logit_new)
    dist_old = torch.distributions.Categorical(logits=logit_old)
    logp_new = dist_new.log_prob(action)
    logp_old = dist_old.log_prob(action)
    if entropy_bonus:
        ent = dist_new.entropy()
        if ent.shape != weight.shape:  # MARL: entropy [B, A]
            ent = ent.mean(dim=1)
        entropy_loss = (ent * weight).mean()
    else:
        entropy_loss = torch.tensor(0.0)
    ratio = torch.exp(logp_new - logp_old)
    if ratio.shape != adv.shape:
        ratio = ratio.mean(dim=1)
    surr1 = ratio * adv
    surr2 = ratio.clamp(1 - clip_ratio, 1 + clip_ratio) * adv
    if dual_clip is not None:
        assert dual_clip > 1.0, f"dual_clip must be > 1.0, got {dual_clip}"
        clipped = torch.min(surr1, surr2)
        # dual clip only bites when adv < 0
        policy_loss = -(torch.where(adv < 0, torch.max(clipped, dual_clip * adv), clipped) * weight).mean()
    else:
        policy_loss = (-torch.min(surr1, surr2) * weight).mean()
    with torch.no_grad():
        approx_kl = (logp_old - logp_new).mean().item()
        clipfrac = ((ratio - 1.0).abs() > clip_ratio).float().mean().item()
    if logit_pretrained is not None:
        logp_pre = torch.distributions.Categorical(logits=logit_pretrained).log_prob(action)
        kl_div = calculate_kl_div(logp_new - logp_pre, kl_type)
    else:
        kl_div = torch.zeros((), dtype=policy_loss.dtype, device=policy_loss.device)
    return ppo_policy_loss(policy_loss, entropy_loss, kl_div), ppo_info(approx_kl, clipfrac)


def ppo_value_error(data: namedtuple, clip_ratio: float = 0.2, use_value_clip: bool = True) -> torch.Tensor:
    """0.5 * (clipped) squared error towards the return target."""
    value_new, value_old, return_, weight = data
    if weight is None:
        weight = torch.ones_like(value_old)
    if use_value_clip:
        value_clip = value_old + (value_new - value_old).clamp(-clip_ratio, clip_ratio)
        v1 = (return_ - value_new).pow(2)
        v2 = (return_ - value_clip).pow(2)
        return 0.5 * (torch.max(v1, v2) * weight).mean()
    return 0.5 * ((return_ - value_new).pow(2) * weight).mean()


def ppo_error(
    data: namedtuple,
    clip_ratio: float = 0.2,
    use_value_clip: bool = True,
    dual_clip: Optional[float] = None,
    kl_type: str = 'k1',
) -> Tuple[namedtuple, namedtuple]:
    """Combined PPO loss for discrete actions. See ppo_data for fields."""
    from ding.ops import dispatch
    logit_new, logit_old, action, value_new, value_old, adv, return_, weight = data[:8]
    logit_pretrained = data[8] if len(data) > 8 else None
    if (
        dual_clip is None and logit_pretrained is None and isinstance(logit_new, torch.Tensor)
        and logit_new.dim() == 2 and action.dim() == 1 and logit_new.dtype == torch.float32
        and dispatch.use_hip_autograd(logit_new)
    ):
        policy_loss, value_loss, entropy_loss, approx_kl, clipfrac = dispatch.fused_ppo_error(
            logit_new, logit_old.detach(), action, value_new, value_old.detach(), adv.detach(), return_.detach(),
            None if weight is None else weight.detach().float(), clip_ratio, use_value_clip
        )
        kl_div = torch.zeros((), dtype=policy_loss.dtype, device=policy_loss.device)
        return ppo_loss(policy_loss, value_loss, entropy_loss, kl_div), \
            ppo_info(float(approx_kl), float(clipfrac))
    pol, info = ppo_policy_error(
        ppo_policy_data(logit_new, logit_old, action, adv, weight, logit_pretrained), clip_ratio, dual_clip,
        kl_type=kl_type
    )
    v_loss = ppo_value_error(ppo_value_data(value_new, value_old, return_, weight), clip_ratio, use_value_clip)
    return ppo_loss(pol.policy_loss, v_loss, pol.entropy_loss, pol.kl_div), info


def _gaussian_dist(mu_sigma: dict):
    return torch.distributions.Independent(torch.distributions.Normal(mu_sigma['mu'], mu_sigma['sigma']), 1)


def ppo_policy_error_continuous(
    data: namedtuple,
    clip_ratio: float = 0.2,
    dual_clip: Optional[float] = None,
) -> Tuple[namedtuple, namedtuple]:
    mu_sigma_new, mu_sigma_old, action, adv, weight = data
    if weight is None:
        weight = torch.ones_like(adv)
    dist_new = _gaussian_dist(mu_sigma_new)
    dist_old = _gaussian_dist(mu_sigma_old)
    logp_new = dist_new.log_prob(action)
    logp_old = dist_old.log_prob(action)
    # multi-agent continuous: log_prob carries a trailing agent dim that
    # adv/weight ([B]) lack — broadcast them across agents
    while adv.dim() < logp_new.dim():
        adv = adv.unsqueeze(-1)
    while weight.dim() < logp_new.dim():
        weight = weight.unsqueeze(-1)
    entropy_loss = (dist_new.entropy() * weight).mean()
    ratio = torch.exp(logp_new - logp_old)
    surr1 = ratio * adv
    surr2 = ratio.clamp(1 - clip_ratio, 1 + clip_ratio) * adv
    if dual_clip is not None:
        assert dual_clip > 1.0
        clipped = torch.min(surr1, surr2)
        policy_loss = -(torch.where(adv < 0, torch.max(clipped, dual_clip * adv), clipped) * weight).mean()
    else:
        policy_loss = (-torch.min(surr1, surr2) * weight).mean()
    with torch.no_grad():
        approx_kl = (logp_old - logp_new).mean().item()
        clipfrac = ((ratio - 1.0).abs() > clip_ratio).float().mean().item()
    kl_div = torch.zeros((), device=policy_loss.device)
    return ppo_policy_loss(policy_loss, entropy_loss, kl_div), ppo_info(approx_kl, clipfrac)


def ppo_error_continuous(
    data: namedtuple,
    clip_ratio: float = 0.2,
    use_value_clip: bool = True,
    dual_clip: Optional[float] = None,
) -> Tuple[namedtuple, namedtuple]:
    # accept both ppo_data (9 fields, trailing logit_pretrained) and
    # ppo_data_continuous (8 fields)
    mu_sigma_new, mu_sigma_old, action, value_new, value_old, adv, return_, weight = tuple(data)[:8]
    pol, info = ppo_policy_error_continuous(
        ppo_policy_data_continuous(mu_sigma_new, mu_sigma_old, action, adv, weight), clip_ratio, dual_clip
    )
    v_loss = ppo_value_error(ppo_value_data(value_new, value_old, return_, weight), clip_ratio, use_value_clip)
    return ppo_loss(pol.policy_loss, v_loss, pol.entropy_loss, pol.kl_div), info
