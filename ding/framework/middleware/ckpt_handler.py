"""Checkpoint middleware.

Parity: reference ding/framework/middleware/ckpt_handler.py (CkptSaver:14):
periodic / best-eval / final saves of policy.learn_mode.state_dict() to
<exp_name>/ckpt/*.pth.tar.
"""
import os
from typing import Optional, Union

import numpy as np

from ding.policy import Policy
from ding.utils import save_file
from ..context import OnlineRLContext, OfflineRLContext


class CkptSaver:

    def __init__(self, policy: Policy, save_dir: str, train_freq: Optional[int] = None, save_finish: bool = True):
        # accept a Policy (use its learn_mode view) or any mode view exposing state_dict
        if hasattr(policy, 'learn_mode'):
            policy = policy.learn_mode
        self.policy = policy
        if save_dir.endswith('ckpt'):
            self.prefix = save_dir
        else:
            self.prefix = os.path.join(save_dir, 'ckpt')
        os.makedirs(self.prefix, exist_ok=True)
        self.last_save_iter = 0
        self.max_eval_value = -np.inf
        self.train_freq = train_freq
        self.save_finish = save_finish

    def __call__(self, ctx: Union[OnlineRLContext, OfflineRLContext]) -> None:
        if self.train_freq and ctx.train_iter > 0:
            if ctx.train_iter == 1 or ctx.train_iter - self.last_save_iter >= self.train_freq:
                save_file(
                    os.path.join(self.prefix, f"iteration_{ctx.train_iter}.pth.tar"),
                    self.policy.state_dict()
                )
                self.last_save_iter = ctx.train_iter
        if not np.isinf(ctx.eval_value) and ctx.eval_value > self.max_eval_value:
            save_file(os.path.join(self.prefix, "eval.pth.tar"), self.policy.state_dict())
            self.max_eval_value = ctx.eval_value
        from ding.framework import task as _task
        if _task.finish and self.save_finish:
            save_file(os.path.join(self.prefix, "final.pth.tar"), self.policy.state_dict())
