"""Dataset-metric evaluator: evaluate a policy against an objective metric
over a dataloader (supervised eval, e.g. procedure cloning / BC accuracy)
instead of env interaction.

Parity: reference ding/worker/collector/metric_serial_evaluator.py
(IMetric:13, MetricSerialEvaluator:36).
"""
from abc import ABC, abstractmethod
from typing import Any, Callable, List, Optional, Tuple

import numpy as np
import torch

from ding.torch_utils import to_tensor
from ding.utils import EasyDict, SERIAL_EVALUATOR_REGISTRY, deep_merge_dicts


class IMetric(ABC):

    @abstractmethod
    def eval(self, inputs: Any, label: Any) -> dict:
        raise NotImplementedError

    @abstractmethod
    def reduce_mean(self, inputs: List[Any]) -> Any:
        raise NotImplementedError

    @abstractmethod
    def gt(self, metric1: Any, metric2: Any) -> bool:
        """metric1 >= metric2 (metric2 None -> True)."""
        raise NotImplementedError


@SERIAL_EVALUATOR_REGISTRY.register('metric')
class MetricSerialEvaluator:
    """``env`` is a ``(dataloader, metric)`` pair; ``eval`` runs the policy
    over the dataloader and reduces the metric, checkpointing on a new best
    and stopping once ``stop_value`` is reached."""

    config = dict(type='metric', eval_freq=50, stop_value=1.0, multi_gpu=False)

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(cls.config)

    def __init__(
        self,
        cfg: EasyDict,
        env: Tuple[Any, IMetric] = None,
        policy=None,
        tb_logger=None,
        exp_name: str = 'default_experiment',
        instance_name: str = 'evaluator',
    ):
        self._cfg = deep_merge_dicts(self.default_config(), cfg or EasyDict({}))
        self._exp_name = exp_name
        self._instance_name = instance_name
        self._stop_value = self._cfg.stop_value
        self._last_eval_iter = -1
        self._max_avg_eval_result = None
        self.reset(policy, env)

    def reset_env(self, _env: Optional[Tuple[Any, IMetric]] = None) -> None:
        if _env is not None:
            self._dataloader, self._metric = _env

    def reset_policy(self, _policy=None) -> None:
        if _policy is not None:
            self._policy = _policy
        self._policy.reset()

    def reset(self, _policy=None, _env: Optional[Tuple[Any, IMetric]] = None) -> None:
        if _env is not None:
            self.reset_env(_env)
        if _policy is not None:
            self.reset_policy(_policy)
        self._max_avg_eval_result = None
        self._last_eval_iter = -1

    def close(self) -> None:
        pass

    def should_eval(self, train_iter: int) -> bool:
        if train_iter == self._last_eval_iter:
            return False
        if (train_iter - self._last_eval_iter) < self._cfg.eval_freq and train_iter != 0:
            return False
        self._last_eval_iter = train_iter
        return True

    def eval(
        self,
        save_ckpt_fn: Callable = None,
        train_iter: int = -1,
        envstep: int = -1,
    ) -> Tuple[bool, Any]:
        self._policy.reset()
        eval_results = []
        for batch_data in self._dataloader:
            inputs, label = to_tensor(batch_data)
            policy_output = self._policy.forward(inputs)
            eval_results.append(self._metric.eval(policy_output, label))
        avg_eval_result = self._metric.reduce_mean(eval_results)
        if self._cfg.multi_gpu and torch.distributed.is_initialized():
            for k in avg_eval_result.keys():
                t = torch.FloatTensor([avg_eval_result[k]])
                torch.distributed.all_reduce(t)
                avg_eval_result[k] = (t / torch.distributed.get_world_size()).item()
        if self._metric.gt(avg_eval_result, self._max_avg_eval_result):
            if save_ckpt_fn:
                save_ckpt_fn('ckpt_best.pth.tar')
            self._max_avg_eval_result = avg_eval_result
        stop_flag = self._metric.gt(avg_eval_result, self._stop_value) and train_iter > 0
        return stop_flag, avg_eval_result
