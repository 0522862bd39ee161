"""Learner aggregator: merge learn info / gradients across learners with
reconnect-on-failure.

Parity: reference ding/worker/adapter/learner_aggregator.py:55.
"""
import logging
import time
from typing import Any, Dict, List, Optional

import numpy as np

logger = logging.getLogger('ding')


class LearnerAggregator:
    """Aggregates learn_info dicts from several learners; tolerates dead
    learners (reconnect hook) and exposes merged metrics."""

    def __init__(self, learner_getters: Optional[List] = None, reconnect_fn=None):
        self._getters = learner_getters or []
        self._reconnect_fn = reconnect_fn
        self._dead = set()

    def register_learner(self, getter) -> None:
        self._getters.append(getter)

    def merge_info(self) -> Dict[str, Any]:
        infos = []
        for i, g in enumerate(self._getters):
            if i in self._dead:
                if self._reconnect_fn is not None and self._reconnect_fn(i):
                    self._dead.discard(i)
                else:
                    continue
            try:
                infos.append(g())
            except Exception as e:
                logger.warning(f"learner {i} unreachable: {e}")
                self._dead.add(i)
        if not infos:
            return {}
        merged: Dict[str, Any] = {}
        keys = set().union(*[set(d.keys()) for d in infos])
        for k in keys:
            vals = [d[k] for d in infos if k in d]
            if all(isinstance(v, (int, float)) for v in vals):
                merged[k] = float(np.mean(vals))
            else:
                merged[k] = vals[-1]
        merged['learner_num'] = len(infos)
        merged['dead_learner_num'] = len(self._dead)
        return merged

    def merge_grads(self, grads_list: List[List]) -> List:
        """Average aligned gradient lists (slow CPU path; the RCCL bucketed
        reducer in ding/parallel is the fast lane)."""
        import torch
        n = len(grads_list)
        return [torch.stack([g[i] for g in grads_list]).mean(0) for i in range(len(grads_list[0]))]
