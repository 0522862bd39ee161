"""UPGO (upgoing policy update) returns and loss.

Parity: reference ding/rl_utils/upgo.py (upgo_returns:46, upgo_loss:77,
tb_cross_entropy:7).
"""
import torch
import torch.nn.functional as F

from .td import generalized_lambda_returns


def tb_cross_entropy(logit: torch.Tensor, label: torch.Tensor, mask: torch.Tensor = None) -> torch.Tensor:
    """Time-batched cross-entropy: logit [T,B,...,N], label [T,B,...].

    Returns -CE per (t, b), averaged over any trailing dims.
    """
    assert len(label.shape) >= 2
    T, B = label.shape[:2]
    flat_logit = logit.reshape(-1, logit.shape[-1])
    flat_label = label.reshape(-1)
    ce = -F.cross_entropy(flat_logit, flat_label, reduction='none')
    if mask is not None:
        ce = ce * mask.reshape(-1)
    ce = ce.reshape(T, B, -1)
    return ce.mean(dim=2)


def upgo_returns(rewards: torch.Tensor, bootstrap_values: torch.Tensor) -> torch.Tensor:
    """Lambda-return with per-step lambda in {0,1}: trace continues when
    r_{t+1} + V_{t+2} >= V_{t+1}. rewards [T,B], bootstrap_values [T+1,B]."""
    lambdas = (rewards + bootstrap_values[1:]) >= bootstrap_values[:-1]
    lambdas = torch.cat([lambdas[1:], torch.ones_like(lambdas[-1:])], dim=0).float()
    return generalized_lambda_returns(bootstrap_values, rewards, 1.0, lambdas)


def upgo_loss(
    target_output: torch.Tensor,
    rhos: torch.Tensor,
    action: torch.Tensor,
    rewards: torch.Tensor,
    bootstrap_values: torch.Tensor,
    mask=None,
) -> torch.Tensor:
    """IS-weighted policy-gradient loss on UPGO advantages.

    target_output [T,B,N]; rhos/action/rewards [T,B]; bootstrap_values [T+1,B].
    """
    with torch.no_grad():
        returns = upgo_returns(rewards, bootstrap_values)
        advantages = rhos * (returns - bootstrap_values[:-1])
    metric = tb_cross_entropy(target_output, action, mask)
    assert metric.shape == action.shape[:2]
    return -(advantages * metric).mean()
