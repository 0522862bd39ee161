"""Policy ABC with learn/collect/eval mode views and RCCL data-parallel
support.

Parity: reference ding/policy/base_policy.py (Policy:14, mode views:292-362,
multi-GPU init:167-199, sync_gradients:415-460, CommandModePolicy:750).

MI355X distributed design: instead of per-parameter async all-reduce hooks,
``multi_gpu`` policies attach a GradBucketAllReducer (ding/parallel) that
packs grads into ~25 MB flat buckets reduced on a dedicated HIP stream
overlapping backward (see its docstring for xGMI sizing rationale).
"""
import copy
from abc import ABC, abstractmethod
from collections import namedtuple
from typing import Any, Dict, List, Optional, Union

import torch

from ding.model import create_model
from ding.utils import (
    POLICY_REGISTRY, EasyDict, deep_merge_dicts, import_module, allreduce_data, broadcast_object_list, get_rank,
    get_world_size, is_dist_initialized,
)


class Policy(ABC):

    learn_function = namedtuple(
        'learn_function', [
            'forward', 'reset', 'info', 'monitor_vars', 'get_attribute', 'set_attribute', 'state_dict',
            'load_state_dict'
        ]
    )
    collect_function = namedtuple(
        'collect_function', [
            'forward', 'process_transition', 'get_train_sample', 'reset', 'get_attribute', 'set_attribute',
            'state_dict', 'load_state_dict'
        ]
    )
    eval_function = namedtuple(
        'eval_function', ['forward', 'reset', 'get_attribute', 'set_attribute', 'state_dict', 'load_state_dict']
    )
    total_field = set(['learn', 'collect', 'eval'])
    config = dict(
        on_policy=False,
        cuda=False,
        multi_gpu=False,
        bp_update_sync=True,
        traj_len_inf=False,
        model=dict(),
    )

    @classmethod
    def default_config(cls) -> EasyDict:
        base = {}
        for klass in reversed(cls.__mro__):
            if hasattr(klass, 'config'):
                base = deep_merge_dicts(base, klass.config)
        cfg = EasyDict(copy.deepcopy(base))
        cfg.cfg_type = cls.__name__ + 'Dict'
        return cfg

    def __init__(
        self,
        cfg: EasyDict,
        model: Optional[torch.nn.Module] = None,
        enable_field: Optional[List[str]] = None,
    ):
        self._cfg = cfg
        self._on_policy = self._cfg.on_policy
        if enable_field is None:
            self._enable_field = self.total_field
        else:
            self._enable_field = enable_field
        assert set(self._enable_field).issubset(self.total_field), self._enable_field

        if len(set(self._enable_field).intersection(set(['learn', 'collect', 'eval']))) > 0:
            model = self._create_model(cfg, model)
            self._cuda = cfg.cuda and torch.cuda.is_available()
            if self._cuda:
                torch.cuda.set_device(get_rank() % max(1, torch.cuda.device_count()))
                model.cuda()
            if len(set(self._enable_field).intersection(set(['learn']))) > 0:
                multi_gpu = self._cfg.multi_gpu
                self._rank = get_rank() if multi_gpu else 0
                if multi_gpu:
                    bp_update_sync = self._cfg.bp_update_sync
                    self._bp_update_sync = bp_update_sync
                    self._init_multi_gpu_setting(model, bp_update_sync)
            else:
                self._rank = 0
            self._model = model
            self._device = 'cuda:{}'.format(torch.cuda.current_device()) if self._cuda else 'cpu'
        else:
            self._cuda = False
            self._rank = 0
            self._device = 'cpu'

        for field in self._enable_field:
            getattr(self, '_init_' + field)()

    def _init_multi_gpu_setting(self, model: torch.nn.Module, bp_update_sync: bool) -> None:
        from ding.parallel import GradBucketAllReducer
        self._grad_reducer = GradBucketAllReducer(model, async_overlap=not bp_update_sync)
        if is_dist_initialized():
            self._grad_reducer.broadcast_params(src=0)

    def sync_gradients(self, model: torch.nn.Module) -> None:
        """All-reduce grads across DP ranks (bucketed flat messages)."""
        if not is_dist_initialized():
            return
        if hasattr(self, '_grad_reducer') and self._grad_reducer.model is model:
            self._grad_reducer.sync()
        else:
            from ding.parallel import sync_gradients_flat
            sync_gradients_flat(model)

    def _create_model(self, cfg: EasyDict, model: Optional[torch.nn.Module] = None) -> torch.nn.Module:
        if model is not None:
            return model
        model_cfg = cfg.model
        if 'type' not in model_cfg:
            m_type, import_names = self.default_model()
            model_cfg.type = m_type
            model_cfg.import_names = import_names
        return create_model(model_cfg)

    @abstractmethod
    def _init_learn(self) -> None:
        raise NotImplementedError

    @abstractmethod
    def _init_collect(self) -> None:
        raise NotImplementedError

    @abstractmethod
    def _init_eval(self) -> None:
        raise NotImplementedError

    # --------------------------------------------------------------- views
    @property
    def learn_mode(self) -> 'Policy.learn_function':
        return Policy.learn_function(
            self._forward_learn,
            self._reset_learn,
            self.__repr__,
            self._monitor_vars_learn,
            self._get_attribute,
            self._set_attribute,
            self._state_dict_learn,
            self._load_state_dict_learn,
        )

    @property
    def collect_mode(self) -> 'Policy.collect_function':
        return Policy.collect_function(
            self._forward_collect,
            self._process_transition,
            self._get_train_sample,
            self._reset_collect,
            self._get_attribute,
            self._set_attribute,
            self._state_dict_collect,
            self._load_state_dict_collect,
        )

    @property
    def eval_mode(self) -> 'Policy.eval_function':
        return Policy.eval_function(
            self._forward_eval,
            self._reset_eval,
            self._get_attribute,
            self._set_attribute,
            self._state_dict_eval,
            self._load_state_dict_eval,
        )

    def _set_attribute(self, name: str, value: Any) -> None:
        setattr(self, '_' + name, value)

    def _get_attribute(self, name: str) -> Any:
        if hasattr(self, '_get_' + name):
            return getattr(self, '_get_' + name)()
        if hasattr(self, '_' + name):
            return getattr(self, '_' + name)
        raise NotImplementedError(name)

    def __repr__(self) -> str:
        return "DI-engine-MI355X policy: {}".format(type(self).__name__)

    # ------------------------------------------------------------ defaults
    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        raise NotImplementedError

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        raise NotImplementedError

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        raise NotImplementedError

    def _reset_learn(self, data_id: Optional[List[int]] = None) -> None:
        pass

    def _reset_collect(self, data_id: Optional[List[int]] = None) -> None:
        pass

    def _reset_eval(self, data_id: Optional[List[int]] = None) -> None:
        pass

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss']

    def _process_transition(self, obs: Any, policy_output: Dict[str, Any], timestep: namedtuple) -> Dict[str, Any]:
        raise NotImplementedError

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        raise NotImplementedError

    def _state_dict_learn(self) -> Dict[str, Any]:
        state = {'model': self._model.state_dict()}
        if hasattr(self, '_optimizer'):
            state['optimizer'] = self._optimizer.state_dict()
        if hasattr(self, '_target_model') and hasattr(self._target_model, 'state_dict'):
            state['target_model'] = self._target_model.state_dict()
        return state

    def _load_state_dict_learn(self, state_dict: Dict[str, Any]) -> None:
        self._model.load_state_dict(state_dict['model'])
        if 'optimizer' in state_dict and hasattr(self, '_optimizer'):
            self._optimizer.load_state_dict(state_dict['optimizer'])
        if 'target_model' in state_dict and hasattr(self, '_target_model'):
            self._target_model.load_state_dict(state_dict['target_model'])

    def _state_dict_collect(self) -> Dict[str, Any]:
        return {'model': self._collect_model.state_dict() if hasattr(self, '_collect_model') else self._model.state_dict()}

    def _load_state_dict_collect(self, state_dict: Dict[str, Any]) -> None:
        if hasattr(self, '_collect_model'):
            self._collect_model.load_state_dict(state_dict['model'], strict=True)
        else:
            self._model.load_state_dict(state_dict['model'], strict=True)

    def _state_dict_eval(self) -> Dict[str, Any]:
        return {'model': self._eval_model.state_dict() if hasattr(self, '_eval_model') else self._model.state_dict()}

    def _load_state_dict_eval(self, state_dict: Dict[str, Any]) -> None:
        if hasattr(self, '_eval_model'):
            self._eval_model.load_state_dict(state_dict['model'], strict=True)
        else:
            self._model.load_state_dict(state_dict['model'], strict=True)

    def default_model(self) -> Union[tuple, None]:
        raise NotImplementedError

    # ---------------------------------------------------------------- misc
    @property
    def cfg(self) -> EasyDict:
        return self._cfg

    def _get_batch_size(self) -> Union[int, Dict[str, int]]:
        if 'learn' in self._enable_field:
            return self._cfg.learn.batch_size
        raise NotImplementedError

    def _get_n_sample(self):
        return self._cfg.collect.get('n_sample', None)

    def _get_n_episode(self):
        return self._cfg.collect.get('n_episode', None)

    def _get_on_policy(self):
        return self._on_policy

    def _get_priority(self):
        if hasattr(self, '_priority'):
            return self._priority
        return self._cfg.get('priority', False)


class CommandModePolicy(Policy):
    """Policy + command mode: per-iteration hyperparameter schedule info used
    by serial/parallel commanders (e.g. eps)."""

    command_function = namedtuple('command_function', ['get_setting_learn', 'get_setting_collect', 'get_setting_eval'])
    total_field = set(['learn', 'collect', 'eval', 'command'])

    @property
    def command_mode(self) -> 'CommandModePolicy.command_function':
        return CommandModePolicy.command_function(
            self._get_setting_learn, self._get_setting_collect, self._get_setting_eval
        )

    @abstractmethod
    def _init_command(self) -> None:
        raise NotImplementedError

    @abstractmethod
    def _get_setting_learn(self, command_info: dict) -> dict:
        raise NotImplementedError

    @abstractmethod
    def _get_setting_collect(self, command_info: dict) -> dict:
        raise NotImplementedError

    @abstractmethod
    def _get_setting_eval(self, command_info: dict) -> dict:
        raise NotImplementedError


def create_policy(cfg: EasyDict, model: Optional[torch.nn.Module] = None, **kwargs) -> Policy:
    import_module(cfg.get('import_names', []))
    import ding.policy  # ensure registry population
    return POLICY_REGISTRY.build(cfg.type, cfg=cfg, model=model, **kwargs)


def get_policy_cls(cfg: EasyDict) -> type:
    import_module(cfg.get('import_names', []))
    import ding.policy
    return POLICY_REGISTRY.get(cfg.type)
