"""Retrace(lambda) Q targets (Safe and efficient off-policy RL, Munos 2016).

Parity: reference ding/rl_utils/retrace.py:7 compute_q_retraces (ACER).
"""
import torch


def compute_q_retraces(
    q_values: torch.Tensor,
    v_pred: torch.Tensor,
    rewards: torch.Tensor,
    actions: torch.Tensor,
    weights: torch.Tensor,
    ratio: torch.Tensor,
    gamma: float = 0.9,
) -> torch.Tensor:
    """Reverse recursion:
      Qret_T = V_T
      Qret_t = r_t + gamma * w_t * carry_{t+1}
      carry_t = min(1, ratio_t(a_t)) (Qret_t - Q_t(a_t)) + V_t

    Shapes: q_values [T+1,B,N], v_pred [T+1,B,1], rewards [T,B],
    actions [T,B], weights [T,B] (1-done style masks), ratio [T,B,N].
    Returns q_retraces [T+1,B,1]; forward-only (no grad needed).
    """
    T = rewards.shape[0]
    rewards = rewards.unsqueeze(-1)
    actions = actions.unsqueeze(-1)
    weights = weights.unsqueeze(-1)
    q_retraces = torch.zeros_like(v_pred)  # [T+1, B, 1]
    q_retraces[-1] = v_pred[-1]
    carry = v_pred[-1]
    q_gather = torch.zeros_like(v_pred)
    q_gather[0:-1] = q_values[0:-1].gather(-1, actions)
    ratio_gather = ratio.gather(-1, actions)  # [T, B, 1]
    for t in range(T - 1, -1, -1):
        q_retraces[t] = rewards[t] + gamma * weights[t] * carry
        carry = ratio_gather[t].clamp(max=1.0) * (q_retraces[t] - q_gather[t]) + v_pred[t]
    return q_retraces
