"""ACER policy/value losses with truncated IS + bias correction.

Parity: reference ding/rl_utils/acer.py (acer_policy_error,
acer_value_error, acer_trust_region_update).
"""
from typing import List, Tuple

import torch

EPS = 1e-8


def acer_policy_error(
    q_values: torch.Tensor,
    q_retraces: torch.Tensor,
    v_pred: torch.Tensor,
    target_logit: torch.Tensor,
    actions: torch.Tensor,
    ratio: torch.Tensor,
    c_clip_ratio: float = 10.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Truncated importance-sampled PG + bias-correction term.

    q_values/target_logit/ratio: [T,B,N]; q_retraces/v_pred: [T,B,1];
    actions: [T,B]. Returns (actor_loss, bc_loss) each [T,B,1].
    """
    actions = actions.unsqueeze(-1)
    with torch.no_grad():
        adv_retraces = q_retraces - v_pred  # [T,B,1]
        adv_bc = (q_values - v_pred).detach()  # [T,B,N]
    logp = target_logit.gather(-1, actions)  # [T,B,1]
    ratio_taken = ratio.gather(-1, actions)
    actor_loss = -ratio_taken.clamp(max=c_clip_ratio) * logp * adv_retraces
    # bias correction over all actions under the target policy
    pi = torch.exp(target_logit)
    bc_weight = (1 - c_clip_ratio / (ratio + EPS)).clamp(min=0) * pi
    bc_loss = -(bc_weight.detach() * target_logit * adv_bc).sum(-1, keepdim=True)
    return actor_loss, bc_loss


def acer_value_error(q_values: torch.Tensor, q_retraces: torch.Tensor, actions: torch.Tensor) -> torch.Tensor:
    """0.5 * (Qret - Q(a))^2 per step; shapes as in acer_policy_error."""
    actions = actions.unsqueeze(-1)
    q_taken = q_values.gather(-1, actions)
    return 0.5 * (q_retraces.detach() - q_taken).pow(2)


def acer_trust_region_update(
    actor_gradients: List[torch.Tensor],
    target_logit: torch.Tensor,
    avg_logit: torch.Tensor,
    trust_region_value: float,
) -> List[torch.Tensor]:
    """Project the actor gradient so KL(avg || target) stays bounded."""
    with torch.no_grad():
        kl_grad = -torch.exp(avg_logit)  # d/dlogit KL(avg, target) (up to const)
        updates = []
        for g in actor_gradients:
            scale = (kl_grad * g).sum(-1, keepdim=True) - trust_region_value
            scale = scale.clamp(min=0) / (kl_grad * kl_grad).sum(-1, keepdim=True).clamp(min=EPS)
            updates.append(g - scale * kl_grad)
    return updates
