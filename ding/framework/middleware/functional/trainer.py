"""Trainer middleware.

Parity: reference ding/framework/middleware/functional/trainer.py (trainer:10,
multistep_trainer:57).
"""
import logging
from typing import Callable

from ding.policy import Policy
from ...context import OnlineRLContext, OfflineRLContext

logger = logging.getLogger('ding')


def trainer(cfg, policy: Policy, log_freq: int = 100) -> Callable:
    """One policy.forward(train_data) per call."""

    def _train(ctx):
        if ctx.train_data is None:
            return
        train_output = policy.forward(ctx.train_data)
        if ctx.train_iter % log_freq == 0:
            loss = train_output.get('total_loss', None)
            if loss is not None:
                logger.info(f"Training: train iter({ctx.train_iter}), loss({loss:.4f})")
        ctx.train_iter += 1
        ctx.train_output = train_output

    return _train


def multistep_trainer(policy: Policy, log_freq: int = 100) -> Callable:
    """For policies whose forward returns a list of per-minibatch outputs
    (e.g. on-policy PPO epochs)."""

    def _train(ctx):
        if ctx.train_data is None:
            return
        train_output = policy.forward(ctx.train_data)
        if isinstance(train_output, dict):
            train_output = [train_output]
        if ctx.train_iter % log_freq == 0 and train_output:
            loss = train_output[-1].get('total_loss', None)
            if loss is not None:
                logger.info(f"Training: train iter({ctx.train_iter}), loss({loss:.4f})")
        ctx.train_iter += len(train_output)
        ctx.train_output = train_output

    return _train
