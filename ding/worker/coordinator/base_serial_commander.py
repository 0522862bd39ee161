"""Per-iteration hyperparameter commander.

Parity: reference ding/worker/coordinator/base_serial_commander.py.
"""
import copy
from collections import namedtuple
from typing import Optional

from ding.utils import EasyDict


class BaseSerialCommander:

    config = dict()

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))

    def __init__(
        self,
        cfg: dict,
        learner=None,
        collector=None,
        evaluator=None,
        replay_buffer=None,
        policy=None,
    ):
        self._cfg = cfg
        self._learner = learner
        self._collector = collector
        self._evaluator = evaluator
        self._replay_buffer = replay_buffer
        self._info = {}
        if policy is not None:
            self.policy = policy

    @property
    def policy(self):
        return self._policy

    @policy.setter
    def policy(self, policy):
        self._policy = policy

    def step(self) -> dict:
        """Return collect_kwargs for this iteration (e.g. scheduled eps)."""
        learn_info = self._learner.learn_info if self._learner else {}
        collector_envstep = self._collector.envstep if self._collector else 0
        self._info.update(learn_info)
        self._info['envstep'] = collector_envstep
        if self._policy is not None:
            return self._policy.get_setting_collect(self._info)
        return {}
