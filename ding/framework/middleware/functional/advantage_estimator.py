"""Advantage estimation middleware.

Parity: reference ding/framework/middleware/functional/advantage_estimator.py
(gae_estimator:18, montecarlo_return_estimator:124).
"""
from typing import Callable, Optional

import torch

from ding.policy import Policy
from ding.rl_utils import gae, gae_data, get_train_sample
from ding.torch_utils import to_device
from ding.utils import EasyDict
from ding.utils.data import ttorch_collate, default_collate, default_decollate
from ...context import OnlineRLContext


def gae_estimator(cfg: EasyDict, policy: Policy, buffer_=None) -> Callable:
    """Compute GAE advantages over ctx.trajectories in one batched call; emit
    ctx.train_data (or push to buffer)."""
    model = policy.get_attribute('model')
    # recompute_adv needs next_obs downstream; without it we can drop it to save memory
    rm_keys = [] if cfg.policy.get('recompute_adv', True) else ['next_obs']

    def _gae(ctx: OnlineRLContext):
        data = ctx.trajectories  # list of transition dicts
        batch = default_collate([dict(d) for d in data], cat_1dim=True)
        if isinstance(batch['obs'], torch.Tensor):
            batch['obs'] = batch['obs'].float()
        with torch.no_grad():
            device = policy.get_attribute('device')
            value = model.forward(to_device(batch['obs'], device), mode='compute_critic')['value'].cpu()
            next_value = model.forward(
                to_device(batch['next_obs'].float() if isinstance(batch['next_obs'], torch.Tensor) else
                          batch['next_obs'], device), mode='compute_critic'
            )['value'].cpu()
        batch['value'] = value
        traj_flag = batch['done'].clone()
        # trajectory boundaries: ends of each env slice
        for idx in ctx.trajectory_end_idx:
            traj_flag[idx] = True
        batch['traj_flag'] = traj_flag
        adv_data = gae_data(
            value.unsqueeze(-1), next_value.unsqueeze(-1), batch['reward'].reshape(-1, 1),
            batch['done'].float().unsqueeze(-1), traj_flag.float().unsqueeze(-1)
        )
        batch['adv'] = gae(adv_data, cfg.policy.collect.discount_factor, cfg.policy.collect.gae_lambda).squeeze(-1)
        for k in rm_keys:
            batch.pop(k, None)
        if buffer_ is None:
            ctx.train_data = default_decollate(batch, ignore=['prev_state'])
            # keep as one collated dict for on-policy epoch training
            ctx.train_data = [dict(d) for d in ctx.train_data]
        else:
            data = default_decollate(batch, ignore=['prev_state'])
            for d in data:
                buffer_.push(EasyDict(d))
        ctx.trajectories = None

    return _gae


def ppof_adv_estimator(policy: Policy) -> Callable:

    def _estimate(ctx: OnlineRLContext):
        data = ctx.trajectories
        batch = default_collate([dict(d) for d in data], cat_1dim=True)
        traj_flag = batch['done'].clone()
        for idx in ctx.trajectory_end_idx:
            traj_flag[idx] = True
        batch['traj_flag'] = traj_flag
        ctx.train_data = batch
        ctx.trajectories = None

    return _estimate


def montecarlo_return_estimator(policy: Policy) -> Callable:
    """Per-episode discounted MC return annotation (PG)."""

    def _estimate(ctx: OnlineRLContext):
        train_data = []
        for episode in ctx.episodes:
            R = 0.0
            gamma = policy.get_attribute('cfg').collect.discount_factor
            for t in reversed(episode):
                R = gamma * R + float(t['reward'].item() if isinstance(t['reward'], torch.Tensor) else t['reward'])
                t['return'] = torch.tensor([R])
            train_data.extend(episode)
        ctx.train_data = [dict(d) for d in train_data]
        ctx.episodes = None

    return _estimate
