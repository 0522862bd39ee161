"""Per-iteration mutable context passed through middleware.

Parity: reference ding/framework/context.py (Context:8, OnlineRLContext:46,
OfflineRLContext:82; kept keys :78,104).
"""
from typing import Any, Dict, List, Optional

import numpy as np


class Context(dict):
    """Attribute-access dict with renew()/keep() lifecycle."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.__dict__['_kept_keys'] = set()
        self.total_step = 0

    def __getattr__(self, key):
        try:
            return self[key]
        except KeyError:
            raise AttributeError(key)

    def __setattr__(self, key, value):
        self[key] = value

    def __delattr__(self, key):
        del self[key]

    def renew(self) -> 'Context':
        """Fresh context of the same type, keeping kept keys."""
        total_step = self.total_step
        ctx = type(self)()
        for key in self._kept_keys:
            if key in self:
                ctx[key] = self[key]
        ctx.total_step = total_step + 1
        return ctx

    def keep(self, *keys: str) -> None:
        for key in keys:
            self._kept_keys.add(key)


class OnlineRLContext(Context):

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        # collect
        self.env_step = 0
        self.env_episode = 0
        self.train_iter = 0
        self.train_data = None
        self.train_output = None
        # collect data
        self.obs = None
        self.action = None
        self.inference_output = None
        self.trajectories = None
        self.episodes = None
        self.trajectory_end_idx = []
        # eval
        self.eval_value = -np.inf
        self.last_eval_iter = -1
        self.last_eval_value = -np.inf
        self.eval_output = None
        self.info_for_logging = {}
        self.keep('env_step', 'env_episode', 'train_iter', 'last_eval_iter', 'last_eval_value')


class OfflineRLContext(Context):

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.trained_env_step = 0
        self.train_epoch = 0
        self.train_iter = 0
        self.train_data = None
        self.train_output = None
        self.eval_value = -np.inf
        self.last_eval_iter = -1
        self.last_eval_value = -np.inf
        self.eval_output = None
        self.info_for_logging = {}
        self.keep('trained_env_step', 'train_iter', 'last_eval_iter', 'last_eval_value')
