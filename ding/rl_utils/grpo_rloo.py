"""GRPO / RLOO policy losses for LLM RLHF-style training.

Parity: reference ding/rl_utils/grpo.py:10 and rloo.py:10. Token-level
shapes: logits [B, S, V], action [B, S].
"""
from collections import namedtuple

import torch

grpo_policy_data = namedtuple('grpo_policy_data', ['logit_new', 'logit_old', 'logit_ref', 'action', 'adv', 'weight'])
grpo_info = namedtuple('grpo_info', ['approx_kl', 'clipfrac'])
grpo_loss_tuple = namedtuple('grpo_loss_tuple', ['policy_loss'])
rloo_policy_data = namedtuple('rloo_policy_data', ['logit_new', 'logit_old', 'action', 'reward', 'weight'])
rloo_info = namedtuple('rloo_info', ['approx_kl', 'clipfrac'])
rloo_loss_tuple = namedtuple('rloo_loss_tuple', ['policy_loss'])


def _token_logp(logit: torch.Tensor, action: torch.Tensor) -> torch.Tensor:
    logp = torch.log_softmax(logit, dim=-1)
    return logp.gather(-1, action.unsqueeze(-1)).squeeze(-1)


def grpo_policy_error(data: namedtuple, clip_ratio: float = 0.2, beta: float = 0.1):
    """Clipped PPO surrogate per token + beta * k3-KL to a reference policy.
    adv is the per-sequence group-normalized advantage [B]."""
    logit_new, logit_old, logit_ref, action, adv, weight = data
    if weight is None:
        weight = torch.ones_like(action, dtype=logit_new.dtype)
    logp_new = _token_logp(logit_new, action)
    logp_old = _token_logp(logit_old, action)
    logp_ref = _token_logp(logit_ref, action)
    ratio = torch.exp(logp_new - logp_old)
    adv_tok = adv.unsqueeze(-1)  # broadcast over sequence
    surr1 = ratio * adv_tok
    surr2 = ratio.clamp(1 - clip_ratio, 1 + clip_ratio) * adv_tok
    per_token = torch.min(surr1, surr2)
    # k3 estimator of KL(pi_new || pi_ref)
    kl = torch.exp(logp_ref - logp_new) - (logp_ref - logp_new) - 1
    per_token = per_token - beta * kl
    policy_loss = -(per_token * weight).sum() / weight.sum().clamp(min=1)
    with torch.no_grad():
        approx_kl = (logp_old - logp_new).mean().item()
        clipfrac = ((ratio - 1.0).abs() > clip_ratio).float().mean().item()
    return grpo_loss_tuple(policy_loss), grpo_info(approx_kl, clipfrac)


def rloo_policy_error(data: namedtuple, clip_ratio: float = 0.2):
    """REINFORCE leave-one-out: reward [K, B_group] -> per-sample baseline is
    the mean of the other K-1 samples; then a clipped PPO surrogate."""
    logit_new, logit_old, action, reward, weight = data
    if weight is None:
        weight = torch.ones_like(action, dtype=logit_new.dtype)
    K = reward.shape[0]
    baseline = (reward.sum(0, keepdim=True) - reward) / max(K - 1, 1)
    adv = (reward - baseline).reshape(-1)  # flatten to [B]
    logp_new = _token_logp(logit_new, action)
    logp_old = _token_logp(logit_old, action)
    ratio = torch.exp(logp_new - logp_old)
    adv_tok = adv.unsqueeze(-1)
    surr1 = ratio * adv_tok
    surr2 = ratio.clamp(1 - clip_ratio, 1 + clip_ratio) * adv_tok
    policy_loss = -(torch.min(surr1, surr2) * weight).sum() / weight.sum().clamp(min=1)
    with torch.no_grad():
        approx_kl = (logp_old - logp_new).mean().item()
        clipfrac = ((ratio - 1.0).abs() > clip_ratio).float().mean().item()
    return rloo_loss_tuple(policy_loss), rloo_info(approx_kl, clipfrac)
