"""Logging middleware.

Parity: reference ding/framework/middleware/functional/logger.py
(online_logger:22, offline_logger:83; wandb variants are no-op offline).
"""
from typing import Callable

import numpy as np

from ding.utils import DistributedWriter
from ...context import OnlineRLContext, OfflineRLContext


def online_logger(record_train_iter: bool = False, train_show_freq: int = 100) -> Callable:
    writer = DistributedWriter.get_instance()
    last_train_show_iter = [-1]

    def _logger(ctx: OnlineRLContext):
        if writer is None:
            return
        if not np.isinf(ctx.eval_value):
            writer.add_scalar('basic/eval_episode_return_mean-env_step', ctx.eval_value, ctx.env_step)
            if record_train_iter:
                writer.add_scalar('basic/eval_episode_return_mean-train_iter', ctx.eval_value, ctx.train_iter)
        if ctx.train_output is not None and ctx.train_iter - last_train_show_iter[0] >= train_show_freq:
            last_train_show_iter[0] = ctx.train_iter
            output = ctx.train_output
            if isinstance(output, (list, tuple)):
                if len(output) == 0:
                    return
                output = output[-1]
            if isinstance(output, (list, tuple)):
                if len(output) == 0:
                    return
                output = output[-1]
            if isinstance(output, dict):
                for k, v in output.items():
                    if k in ('priority', ):
                        continue
                    if np.isscalar(v) or (hasattr(v, 'ndim') and getattr(v, 'ndim', 1) == 0):
                        writer.add_scalar(f'basic/train_{k}-env_step', float(v), ctx.env_step)
                        if record_train_iter:
                            writer.add_scalar(f'basic/train_{k}-train_iter', float(v), ctx.train_iter)

    return _logger


def offline_logger(train_show_freq: int = 100) -> Callable:
    writer = DistributedWriter.get_instance()

    def _logger(ctx: OfflineRLContext):
        if writer is None:
            return
        if not np.isinf(ctx.eval_value):
            writer.add_scalar('basic/eval_episode_return_mean-train_iter', ctx.eval_value, ctx.train_iter)
        if ctx.train_output is not None and isinstance(ctx.train_output, dict):
            for k, v in ctx.train_output.items():
                if np.isscalar(v):
                    writer.add_scalar(f'basic/train_{k}-train_iter', float(v), ctx.train_iter)

    return _logger


def wandb_online_logger(*args, **kwargs) -> Callable:
    """wandb unavailable offline: metrics go to the JSONL writer instead."""
    return online_logger()


def wandb_offline_logger(*args, **kwargs) -> Callable:
    return offline_logger()
