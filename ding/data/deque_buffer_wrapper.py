"""Adapter exposing the middleware DequeBuffer through the legacy replay
buffer interface (sample/push/update/count) so worker-pipeline code can use
it directly.

Parity: reference ding/data/buffer/deque_buffer_wrapper.py
(DequeBufferWrapper:14).
"""
import copy
from typing import Optional

import numpy as np

from ding.data.buffer import DequeBuffer
from ding.data.buffer.middleware import PriorityExperienceReplay, use_time_check
from ding.utils import BUFFER_REGISTRY, EasyDict


@BUFFER_REGISTRY.register('deque_wrapper')
class DequeBufferWrapper:

    config = dict(
        replay_buffer_size=10000,
        max_use=float('inf'),
        train_iter_per_log=100,
        priority=False,
        priority_IS_weight=False,
        priority_power_factor=0.6,
        IS_weight_power_factor=0.4,
        IS_weight_anneal_train_iter=int(1e5),
        priority_max_limit=1000,
    )

    @classmethod
    def default_config(cls) -> EasyDict:
        cfg = EasyDict(copy.deepcopy(cls.config))
        cfg.cfg_type = cls.__name__ + 'Dict'
        return cfg

    def __init__(self, cfg: EasyDict, tb_logger: Optional[object] = None, exp_name: str = 'default_experiment',
                 instance_name: str = 'buffer') -> None:
        self.cfg = cfg
        self.name = f'{instance_name}_iter'
        self.tb_logger = tb_logger
        self.priority_max_limit = cfg.get('priority_max_limit', 1000)
        self.buffer = DequeBuffer(size=cfg.replay_buffer_size)
        self.last_log_train_iter = -1
        self.last_sample_index = None
        self.last_sample_meta = None
        if cfg.get('max_use', float('inf')) != float('inf'):
            self.buffer.use(use_time_check(self.buffer, max_use=cfg.max_use))
        if cfg.get('priority', False):
            self.buffer.use(
                PriorityExperienceReplay(
                    self.buffer,
                    IS_weight=cfg.get('priority_IS_weight', False),
                    priority_power_factor=cfg.get('priority_power_factor', 0.6),
                    IS_weight_power_factor=cfg.get('IS_weight_power_factor', 0.4),
                    IS_weight_anneal_train_iter=cfg.get('IS_weight_anneal_train_iter', int(1e5)),
                )
            )

    def sample(self, size: int, train_iter: int = 0):
        output = self.buffer.sample(size=size, ignore_insufficient=True)
        if not output:
            return None
        meta = [o.meta for o in output]
        if self.cfg.get('priority', False):
            self.last_sample_index = [o.index for o in output]
            self.last_sample_meta = meta
        if self.tb_logger is not None and (self.last_log_train_iter == -1
                                           or train_iter - self.last_log_train_iter >= self.cfg.train_iter_per_log):
            if self.cfg.get('max_use', float('inf')) != float('inf'):
                self.tb_logger.add_scalar(
                    f'{self.name}/use_count_avg', float(np.mean([m.get('use_count', 0) for m in meta])), train_iter
                )
            if self.cfg.get('priority', False):
                ps = [m['priority'] for m in meta]
                self.tb_logger.add_scalar(f'{self.name}/priority_avg', float(np.mean(ps)), train_iter)
                self.tb_logger.add_scalar(f'{self.name}/priority_max', float(np.max(ps)), train_iter)
            self.tb_logger.add_scalar(f'{self.name}/buffer_data_count', self.buffer.count(), train_iter)
            self.last_log_train_iter = train_iter
        data = [o.data for o in output]
        if self.cfg.get('priority_IS_weight', False):
            for d, o in zip(data, output):
                d['IS'] = o.meta['priority_IS']
        return data

    def push(self, data, cur_collector_envstep: int = -1) -> None:
        for d in data:
            meta = {}
            if self.cfg.get('priority', False) and isinstance(d, dict) and 'priority' in d:
                meta['priority'] = d.pop('priority')
            self.buffer.push(d, meta=meta)

    def update(self, meta: dict) -> None:
        """Write back new priorities for the last sampled batch."""
        if not self.cfg.get('priority', False) or self.last_sample_index is None:
            return
        for m, p in zip(self.last_sample_meta, meta['priority']):
            m['priority'] = min(self.priority_max_limit, p)
        for idx, m in zip(self.last_sample_index, self.last_sample_meta):
            self.buffer.update(idx, data=None, meta=m)
        self.last_sample_index = None
        self.last_sample_meta = None

    def count(self) -> int:
        return self.buffer.count()

    def save_data(self, file_name: str) -> None:
        self.buffer.save_data(file_name)

    def load_data(self, file_name: str) -> None:
        self.buffer.load_data(file_name)
