"""Exploration middleware.

Parity: reference ding/framework/middleware/functional/explorer.py
(eps_greedy_handler, eps_greedy_masker).
"""
from typing import Callable

from ding.rl_utils import get_epsilon_greedy_fn
from ding.utils import EasyDict
from ...context import OnlineRLContext


def eps_greedy_handler(cfg: EasyDict) -> Callable:
    """Write the scheduled eps into ctx.collect_kwargs each iteration."""
    eps_cfg = cfg.policy.other.eps
    handle = get_epsilon_greedy_fn(eps_cfg.start, eps_cfg.end, eps_cfg.decay, eps_cfg.type)

    def _eps_greedy(ctx: OnlineRLContext):
        ctx.collect_kwargs = dict(getattr(ctx, 'collect_kwargs', {}) or {})
        ctx.collect_kwargs['eps'] = handle(ctx.env_step)
        yield
        try:
            ctx.collect_kwargs.pop('eps')
        except KeyError:
            pass

    return _eps_greedy


def eps_greedy_masker() -> Callable:
    """Force eps=-1 (pure greedy) for expert/demo collection."""

    def _masker(ctx: OnlineRLContext):
        ctx.collect_kwargs = dict(getattr(ctx, 'collect_kwargs', {}) or {})
        ctx.collect_kwargs['eps'] = -1

    return _masker
