"""Data-enhancement middleware.

Parity: reference ding/framework/middleware/functional/enhancer.py
(reward_estimator:12, her_data_enhancer:74, nstep_reward_enhancer:33).
"""
from typing import Callable

import torch

from ding.utils import EasyDict
from ...context import OnlineRLContext


def reward_estimator(cfg: EasyDict, reward_model) -> Callable:
    """Overwrite rewards in ctx.train_data with the reward model estimate."""

    def _estimate(ctx: OnlineRLContext):
        if ctx.train_data is not None:
            reward_model.estimate(ctx.train_data)

    return _estimate


def her_data_enhancer(cfg: EasyDict, buffer_, her_reward_model) -> Callable:
    """Sample episodes and relabel goals via HER before training."""

    def _enhance(ctx: OnlineRLContext):
        if her_reward_model.episode_size is None:
            size = cfg.policy.learn.batch_size
        else:
            size = her_reward_model.episode_size
        try:
            buffered = buffer_.sample(size)
        except (ValueError, AssertionError):
            ctx.train_data = None
            return
        train_data = []
        for episode in [d.data for d in buffered]:
            train_data.extend(her_reward_model.estimate(episode))
        ctx.train_data = sum(train_data, []) if train_data and isinstance(train_data[0], list) else train_data

    return _enhance


def nstep_reward_enhancer(cfg: EasyDict) -> Callable:
    """Rewrite trajectories in-place with n-step reward/next_obs/done/
    value_gamma (middleware analog of Adder.get_nstep_return_data)."""
    from ding.rl_utils import get_nstep_return_data

    def _enhance(ctx: OnlineRLContext):
        nstep = cfg.policy.nstep
        gamma = cfg.policy.get('discount_factor', cfg.policy.learn.get('discount_factor', 0.99))
        from collections import deque
        # per-env slices to avoid crossing trajectory boundaries
        start = 0
        out = []
        for end in ctx.trajectory_end_idx:
            sl = ctx.trajectories[start:end + 1]
            out.extend(get_nstep_return_data(deque(sl), nstep, gamma=gamma))
            start = end + 1
        ctx.trajectories = out

    return _enhance
