"""Termination middleware.

Parity: reference ding/framework/middleware/functional/termination_checker.py
(termination_checker:12, ddp_termination_checker:31 — rank 0 decides,
broadcast over RCCL each iteration).
"""
from typing import Callable, Optional

import numpy as np
import torch

from ...context import OnlineRLContext, OfflineRLContext


def termination_checker(max_env_step: Optional[int] = None, max_train_iter: Optional[int] = None) -> Callable:
    if max_env_step is None:
        max_env_step = np.inf
    if max_train_iter is None:
        max_train_iter = np.inf

    def _check(ctx):
        from ding.framework import task as _task
        if getattr(ctx, 'env_step', 0) >= max_env_step:
            _task.finish = True
        if getattr(ctx, 'train_iter', 0) >= max_train_iter:
            _task.finish = True

    return _check


def ddp_termination_checker(max_env_step: Optional[int] = None, max_train_iter: Optional[int] = None,
                            rank: int = 0) -> Callable:
    import torch.distributed as dist
    if max_env_step is None:
        max_env_step = np.inf
    if max_train_iter is None:
        max_train_iter = np.inf

    def _check(ctx):
        from ding.framework import task as _task
        if rank == 0:
            if getattr(ctx, 'env_step', 0) >= max_env_step or getattr(ctx, 'train_iter', 0) >= max_train_iter:
                finish = torch.ones(1, dtype=torch.int64)
            else:
                finish = torch.zeros(1, dtype=torch.int64)
        else:
            finish = torch.zeros(1, dtype=torch.int64)
        if torch.cuda.is_available():
            finish = finish.cuda()
        dist.broadcast(finish, 0)
        if finish.item():
            _task.finish = True

    return _check


def epoch_timer() -> Callable:
    """Record per-iteration wall time into ctx.epoch_time."""
    import time

    def _timer(ctx):
        start = time.time()
        yield
        ctx.epoch_time = time.time() - start

    return _timer


def final_ctx_saver(name: str) -> Callable:

    def _save(ctx):
        yield
        from ding.framework import task as _task
        if _task.finish:
            import pickle
            import os
            os.makedirs(name, exist_ok=True)
            with open(os.path.join(name, 'result.pkl'), 'wb') as f:
                final = {
                    k: v
                    for k, v in ctx.items() if np.isscalar(v) or isinstance(v, (str, bool, float, int))
                }
                pickle.dump(final, f)

    return _save
