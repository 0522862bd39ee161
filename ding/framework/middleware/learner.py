"""Learner middleware classes.

Parity: reference ding/framework/middleware/learner.py (OffPolicyLearner:15,
HERLearner:69).
"""
from typing import Callable, List, Optional, Union

from ding.data import Buffer
from ding.policy import Policy
from ding.utils import EasyDict
from ..context import OnlineRLContext
from .functional.data_processor import offpolicy_data_fetcher
from .functional.trainer import trainer


class OffPolicyLearner:
    """update_per_collect x { fetch -> [reward model] -> train }."""

    def __init__(
        self,
        cfg: EasyDict,
        policy: Policy,
        buffer_: Union[Buffer, List[Buffer], dict],
        reward_model=None,
        log_freq: int = 100,
    ) -> None:
        self.cfg = cfg
        self._fetcher = offpolicy_data_fetcher(cfg, buffer_)
        self._trainer = trainer(cfg, policy, log_freq=log_freq)
        if reward_model is not None:
            from .functional.enhancer import reward_estimator
            self._reward_estimator = reward_estimator(cfg, reward_model)
        else:
            self._reward_estimator = None

    def __call__(self, ctx: OnlineRLContext) -> None:
        train_output_queue = []
        for _ in range(self.cfg.policy.learn.update_per_collect):
            fetch_gen = self._fetcher(ctx)
            if fetch_gen is not None and hasattr(fetch_gen, '__next__'):
                try:
                    next(fetch_gen)
                except StopIteration:
                    fetch_gen = None
            if ctx.train_data is None:
                break
            if self._reward_estimator:
                self._reward_estimator(ctx)
            self._trainer(ctx)
            # resume fetcher generator so priority updates flow back
            if fetch_gen is not None:
                try:
                    next(fetch_gen)
                except StopIteration:
                    pass
            train_output_queue.append(ctx.train_output)
        ctx.train_output = train_output_queue


class HERLearner:
    """Learner with hindsight experience replay enhancement."""

    def __init__(self, cfg: EasyDict, policy, buffer_, her_reward_model) -> None:
        self.cfg = cfg
        self._her_reward_model = her_reward_model
        self._buffer = buffer_
        self._policy = policy
        self._trainer = trainer(cfg, policy)

    def __call__(self, ctx: OnlineRLContext) -> None:
        for _ in range(self.cfg.policy.learn.update_per_collect):
            try:
                buffered = self._buffer.sample(self._her_reward_model.episode_size)
            except (ValueError, AssertionError):
                break
            episodes = [d.data for d in buffered]
            new_samples = []
            for ep in episodes:
                new_samples.extend(self._her_reward_model.estimate(ep))
            train_data = []
            for s in new_samples:
                train_data.extend(s) if isinstance(s, list) else train_data.append(s)
            ctx.train_data = train_data
            self._trainer(ctx)

    @property
    def policy(self):
        return self._policy
