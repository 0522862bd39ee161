"""Middleware wrapper measuring per-step wall time.

Parity: reference ding/framework/wrapper/step_timer.py (StepTimer:11).
Wrap any middleware (plain or generator) with ``task.use(StepTimer()(mw))``
to log smoothed per-step cost.
"""
import logging
import time
from collections import defaultdict, deque
from functools import wraps
from types import GeneratorType
from typing import Callable

import numpy as np


class StepTimer:

    def __init__(self, print_per_step: int = 1, smooth_window: int = 10):
        self.print_per_step = print_per_step
        self.records = defaultdict(lambda: deque(maxlen=print_per_step * smooth_window))

    def __call__(self, fn: Callable) -> Callable:
        step_name = getattr(fn, '__name__', type(fn).__name__)

        @wraps(fn)
        def executor(ctx):
            start = time.time()
            cost = 0.0
            g = fn(ctx)
            if isinstance(g, GeneratorType):
                try:
                    next(g)
                except StopIteration:
                    pass
                cost = time.time() - start
                yield
                start = time.time()
                try:
                    next(g)
                except StopIteration:
                    pass
                cost += time.time() - start
            else:
                cost = time.time() - start
            self.records[step_name].append(cost)
            if getattr(ctx, 'total_step', 0) % self.print_per_step == 0:
                logging.info(
                    '[Step Timer] %s: Cost: %.2fms, Mean: %.2fms', step_name, cost * 1000,
                    float(np.mean(self.records[step_name])) * 1000
                )

        return executor
