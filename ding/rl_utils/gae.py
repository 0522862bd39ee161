"""Generalized Advantage Estimation.

Parity: reference ding/rl_utils/gae.py:25 (shapes [T, B], optional
trailing agent dim). GPU inputs dispatch to the HIP reverse-scan kernel
(ding/ops/csrc/scan_ops.hip) — one wavefront lane per batch column instead of
a T-long Python loop of device kernels.
"""
from collections import namedtuple

import torch

gae_data = namedtuple('gae_data', ['value', 'next_value', 'reward', 'done', 'traj_flag'])


def gae(data: namedtuple, gamma: float = 0.99, lambda_: float = 0.97) -> torch.FloatTensor:
    """delta_t = r_t + gamma*(1-done_t)*V_{t+1} - V_t;
    A_t = delta_t + gamma*lambda*(1-traj_flag_t)*A_{t+1} (reverse scan)."""
    from ding.ops import dispatch
    value, next_value, reward, done, traj_flag = data
    if done is None:
        done = torch.zeros_like(reward)
    if traj_flag is None:
        traj_flag = done
    done = done.float()
    traj_flag = traj_flag.float()
    if value.dim() == reward.dim() + 1:  # MARL: value [T,B,A], reward [T,B]
        reward = reward.unsqueeze(-1)
        done = done.unsqueeze(-1)
        traj_flag = traj_flag.unsqueeze(-1)
    next_value = next_value * (1 - done)
    delta = reward + gamma * next_value - value
    factor = gamma * lambda_ * (1 - traj_flag)
    if dispatch.use_hip(delta):
        return dispatch.gae_scan(delta, factor)
    adv = torch.zeros_like(value)
    acc = torch.zeros_like(value[0])
    for t in range(reward.shape[0] - 1, -1, -1):
        acc = delta[t] + factor[t] * acc
        adv[t] = acc
    return adv
