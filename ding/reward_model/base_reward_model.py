"""Reward model ABC + factory.

Parity: reference ding/reward_model/base_reward_model.py.
"""
import copy
from abc import ABC, abstractmethod
from typing import Any, List

from ding.utils import REWARD_MODEL_REGISTRY, EasyDict, import_module


class BaseRewardModel(ABC):

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))

    @abstractmethod
    def estimate(self, data: list) -> List[dict]:
        raise NotImplementedError

    @abstractmethod
    def train(self, data=None) -> None:
        raise NotImplementedError

    @abstractmethod
    def collect_data(self, data) -> None:
        raise NotImplementedError

    @abstractmethod
    def clear_data(self) -> None:
        raise NotImplementedError

    def load_expert_data(self, data) -> None:
        pass

    def reward_deepcopy(self, train_data: list) -> list:
        """Copy transitions so reward rewriting never mutates buffer storage."""
        import torch
        out = []
        for item in train_data:
            new_item = {k: v for k, v in item.items()}
            if isinstance(new_item.get('reward'), torch.Tensor):
                new_item['reward'] = new_item['reward'].clone()
            out.append(new_item)
        return out

    def state_dict(self) -> dict:
        if hasattr(self, 'reward_model'):
            return {'model': self.reward_model.state_dict()}
        return {}

    def load_state_dict(self, d: dict) -> None:
        if hasattr(self, 'reward_model') and 'model' in d:
            self.reward_model.load_state_dict(d['model'])


def get_reward_model_cls(cfg: EasyDict) -> type:
    """Registry lookup without construction (reference base_reward_model.py:140)."""
    from ding.utils import import_module
    import_module(cfg.get('import_names', []))
    return REWARD_MODEL_REGISTRY.get(cfg.type)


def create_reward_model(cfg: EasyDict, device: str = 'cpu', tb_logger=None) -> BaseRewardModel:
    cfg = EasyDict(copy.deepcopy(cfg))
    if 'import_names' in cfg:
        import_module(cfg.import_names)
    if 'reward_model' in cfg:
        cfg = cfg.reward_model
    return REWARD_MODEL_REGISTRY.build(cfg.type, config=cfg, device=device, tb_logger=tb_logger)
