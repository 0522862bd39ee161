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


def _wandb_or_none():
    try:
        import wandb
        return wandb
    except ImportError:
        return None


def wandb_online_logger(
    record_path: str = None,
    cfg=None,
    exp_config=None,
    metric_list: list = None,
    env=None,
    model=None,
    anonymous: bool = False,
    project_name: str = 'di-engine-amd',
    run_name: str = None,
    wandb_sweep: bool = False,
) -> Callable:
    """Per-iteration wandb logging of train losses + eval return (reference
    ding/framework/middleware/functional/logger.py:157 wandb_online_logger).
    Falls back to the JSONL DistributedWriter lane when wandb is not
    importable (offline images)."""
    wandb = _wandb_or_none()
    if wandb is None:
        return online_logger()
    if wandb.run is None:
        wandb.init(
            project=project_name, name=run_name, anonymous='allow' if anonymous else None,
            config=None if exp_config is None else dict(exp_config), reinit=True
        )
    if model is not None and cfg is not None and getattr(cfg, 'gradient_logger', False):
        wandb.watch(model)
    fallback = online_logger()

    def _logger(ctx: "OnlineRLContext"):
        fallback(ctx)
        payload = {'env_step': ctx.env_step, 'train_iter': ctx.train_iter}
        outputs = ctx.train_output if isinstance(ctx.train_output, list) else (
            [ctx.train_output] if isinstance(ctx.train_output, dict) else []
        )
        for out in outputs:
            for k, v in out.items():
                if np.isscalar(v) or (hasattr(v, 'ndim') and getattr(v, 'ndim', 1) == 0):
                    if metric_list is None or k in metric_list:
                        payload[f'train/{k}'] = float(v)
        if ctx.eval_value is not None and not np.isinf(ctx.eval_value):
            payload['eval/episode_return'] = float(ctx.eval_value)
        wandb.log(payload, step=ctx.env_step)

    return _logger


def wandb_offline_logger(
    record_path: str = None,
    cfg=None,
    exp_config=None,
    metric_list: list = None,
    env=None,
    model=None,
    anonymous: bool = False,
    project_name: str = 'di-engine-amd',
    run_name: str = None,
    **kwargs,
) -> Callable:
    """Offline-RL wandb logging (reference logger.py:402); JSONL fallback
    when wandb is absent."""
    wandb = _wandb_or_none()
    if wandb is None:
        return offline_logger()
    if wandb.run is None:
        wandb.init(
            project=project_name, name=run_name, anonymous='allow' if anonymous else None,
            config=None if exp_config is None else dict(exp_config), reinit=True
        )
    fallback = offline_logger()

    def _logger(ctx: "OfflineRLContext"):
        fallback(ctx)
        payload = {'train_iter': ctx.train_iter}
        if isinstance(ctx.train_output, dict):
            for k, v in ctx.train_output.items():
                if np.isscalar(v) and (metric_list is None or k in metric_list):
                    payload[f'train/{k}'] = float(v)
        if ctx.eval_value is not None and not np.isinf(ctx.eval_value):
            payload['eval/episode_return'] = float(ctx.eval_value)
        wandb.log(payload, step=ctx.train_iter)

    return _logger
