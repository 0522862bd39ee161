"""Checkpoint save/load with prefix surgery and resilience decorator.

Parity: reference ding/torch_utils/checkpoint_helper.py (build_checkpoint_helper,
CheckpointHelper, auto_checkpoint, CountVar). Checkpoint format preserved:
dict {'model': state_dict, ...} in .pth.tar files.
"""
import logging
import signal
import traceback
from typing import Callable, Optional

import torch

from ding.utils import read_file, save_file

logger = logging.getLogger('ding')


class CountVar:
    """A mutable int for train-iteration counters shared by hooks."""

    def __init__(self, init_val: int = 0):
        self._val = init_val

    @property
    def val(self) -> int:
        return self._val

    def update(self, val: int) -> None:
        self._val = val

    def add(self, add_num: int) -> None:
        self._val += add_num


def build_checkpoint_helper(cfg=None, rank: int = 0) -> "CheckpointHelper":
    return CheckpointHelper()


class CheckpointHelper:

    def _remove_prefix(self, state_dict: dict, prefix: str = 'module.') -> dict:
        return {k[len(prefix):] if k.startswith(prefix) else k: v for k, v in state_dict.items()}

    def _add_prefix(self, state_dict: dict, prefix: str = 'module.') -> dict:
        return {prefix + k: v for k, v in state_dict.items()}

    def save(
        self,
        path: str,
        model: torch.nn.Module,
        optimizer=None,
        last_iter: Optional[CountVar] = None,
        last_epoch: Optional[CountVar] = None,
        last_frame: Optional[CountVar] = None,
        dataset=None,
        collector_info=None,
        prefix_op: Optional[str] = None,
        prefix: Optional[str] = None,
    ) -> None:
        checkpoint = {}
        state_dict = model.state_dict()
        if prefix_op is not None:
            fn = {'remove': self._remove_prefix, 'add': self._add_prefix}[prefix_op]
            state_dict = fn(state_dict, prefix)
        checkpoint['model'] = state_dict
        if optimizer is not None:
            checkpoint['optimizer'] = optimizer.state_dict()
        if last_iter is not None:
            checkpoint['last_iter'] = last_iter.val
        if last_epoch is not None:
            checkpoint['last_epoch'] = last_epoch.val
        if last_frame is not None:
            checkpoint['last_frame'] = last_frame.val
        if dataset is not None:
            checkpoint['dataset'] = dataset.state_dict()
        if collector_info is not None:
            checkpoint['collector_info'] = collector_info.state_dict()
        save_file(path, checkpoint)
        logger.info(f'save checkpoint in {path}')

    def load(
        self,
        load_path: str,
        model: torch.nn.Module,
        optimizer=None,
        last_iter: Optional[CountVar] = None,
        last_epoch: Optional[CountVar] = None,
        last_frame: Optional[CountVar] = None,
        lr_schduler=None,
        dataset=None,
        collector_info=None,
        prefix_op: Optional[str] = None,
        prefix: Optional[str] = None,
        strict: bool = True,
        logger_prefix: str = '',
        state_dict_mask: list = (),
    ) -> None:
        checkpoint = read_file(load_path)
        state_dict = checkpoint['model']
        if prefix_op is not None:
            fn = {'remove': self._remove_prefix, 'add': self._add_prefix}[prefix_op]
            state_dict = fn(state_dict, prefix)
        if state_dict_mask:
            strict = False
            state_dict = {k: v for k, v in state_dict.items() if not any(k.startswith(m) for m in state_dict_mask)}
        model.load_state_dict(state_dict, strict=strict)
        if optimizer is not None and 'optimizer' in checkpoint:
            optimizer.load_state_dict(checkpoint['optimizer'])
        if last_iter is not None and 'last_iter' in checkpoint:
            last_iter.update(checkpoint['last_iter'])
        if last_epoch is not None and 'last_epoch' in checkpoint:
            last_epoch.update(checkpoint['last_epoch'])
        if dataset is not None and 'dataset' in checkpoint:
            dataset.load_state_dict(checkpoint['dataset'])
        if collector_info is not None and 'collector_info' in checkpoint:
            collector_info.load_state_dict(checkpoint['collector_info'])
        logger.info(f'{logger_prefix}load checkpoint from {load_path}')


def auto_checkpoint(func: Callable) -> Callable:
    """Wrap a train loop: on exception/SIGUSR1 call instance.save_checkpoint()."""

    def wrapper(*args, **kwargs):
        handle = args[0]
        assert hasattr(handle, 'save_checkpoint')
        try:
            return func(*args, **kwargs)
        except BaseException as e:
            handle.save_checkpoint()
            traceback.print_exc()
            raise e

    return wrapper
