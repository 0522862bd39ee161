"""Plateau-based hyperparameter scheduler.

Parity: reference ding/utils/scheduler_helper.py (Scheduler) — adjusts one
policy hyperparameter when a monitored metric plateaus.
"""
from .edict import EasyDict


class Scheduler:

    config = dict(
        schedule_flag=False,
        schedule_mode="reduce",
        factor=0.05,
        change_range=[-1, 1],
        threshold=1e-4,
        optimize_mode="min",
        patience=10,
        cooldown=0,
    )

    def __init__(self, merged_scheduler_config: EasyDict):
        cfg = EasyDict(dict(self.config, **(merged_scheduler_config or {})))
        assert cfg.schedule_mode in ("reduce", "add")
        assert cfg.optimize_mode in ("min", "max")
        self.schedule_mode = cfg.schedule_mode
        self.factor = cfg.factor
        self.change_range = cfg.change_range
        self.threshold = cfg.threshold
        self.optimize_mode = cfg.optimize_mode
        self.patience = cfg.patience
        self.cooldown = cfg.cooldown
        self.cooldown_counter = cfg.cooldown
        self.best = None
        self.bad_epochs_num = 0
        self.last_epoch = -1

    def step(self, metrics: float, param: float) -> float:
        self.last_epoch += 1
        if self.is_better(metrics):
            self.best = metrics
            self.bad_epochs_num = 0
        else:
            self.bad_epochs_num += 1
        if self.in_cooldown:
            self.cooldown_counter -= 1
            self.bad_epochs_num = 0
        if self.bad_epochs_num > self.patience:
            param = self.update_param(param)
            self.cooldown_counter = self.cooldown
            self.bad_epochs_num = 0
        return param

    def update_param(self, param: float) -> float:
        delta = -self.factor if self.schedule_mode == "reduce" else self.factor
        lo, hi = self.change_range
        return min(max(param + delta, lo), hi)

    @property
    def in_cooldown(self) -> bool:
        return self.cooldown_counter > 0

    def is_better(self, metrics: float) -> bool:
        if self.best is None:
            return True
        if self.optimize_mode == "min":
            return metrics < self.best - self.threshold
        return metrics > self.best + self.threshold
