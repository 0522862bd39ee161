"""Layered config compilation.

Parity: reference ding/config/config.py (read_config:180, compile_config:328,
compile_config_parallel:477, save_config:221): deep-merge
default_config(policy/env-manager/buffer/evaluator...) <- user cfg <-
create_cfg, seed it, persist formatted_total_config.
"""
import datetime
import importlib.util
import json
import os
from copy import deepcopy
from typing import Optional, Tuple

import yaml

from ding.utils import EasyDict, deep_merge_dicts

# defaults mirrored from reference config/config.py helper tables
env_cfg_template = dict(
    manager=dict(
        episode_num=float("inf"),
        max_retry=1,
        retry_type='reset',
        auto_reset=True,
        step_timeout=None,
        reset_timeout=None,
        retry_waiting_time=0.1,
        shared_memory=True,
        copy_on_get=True,
    ),
)

main_cfg_template = dict(
    exp_name='default_experiment',
    seed=0,
    env=env_cfg_template,
    policy=dict(),
)


def read_config(path: str) -> Tuple[EasyDict, EasyDict]:
    """Load a .py config module exposing main_config (+ create_config) or a
    yaml file."""
    assert os.path.exists(path), path
    if path.endswith('.py'):
        spec = importlib.util.spec_from_file_location('ding_user_config', path)
        module = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(module)
        main_config = EasyDict(module.main_config)
        create_config = EasyDict(getattr(module, 'create_config', {}))
        return main_config, create_config
    if path.endswith(('.yaml', '.yml')):
        with open(path) as f:
            d = yaml.safe_load(f)
        return EasyDict(d.get('main_config', d)), EasyDict(d.get('create_config', {}))
    raise ValueError(f"unsupported config file: {path}")


def save_config_py(cfg: dict, path: str) -> None:
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    body = json.dumps(_plain(cfg), indent=4, default=repr)
    # json -> python literal syntax
    body = body.replace(': true', ': True').replace(': false', ': False').replace(': null', ': None')
    body = body.replace(': Infinity', ": float('inf')").replace(': -Infinity', ": float('-inf')")
    with open(path, 'w') as f:
        f.write('exp_config = ' + body + '\n')


def save_config_yaml(cfg: dict, path: str) -> None:
    with open(path, 'w') as f:
        yaml.safe_dump(_plain(cfg), f)


def save_config(cfg: dict, path: str, type_: str = 'py', save_formatted: bool = False) -> None:
    if type_ == 'py':
        save_config_py(cfg, path)
    else:
        save_config_yaml(cfg, path)


def _plain(x):
    if isinstance(x, dict):
        return {k: _plain(v) for k, v in x.items()}
    if isinstance(x, (list, tuple)):
        return [_plain(v) for v in x]
    if isinstance(x, (int, float, str, bool, type(None))):
        return x
    return repr(x)


def compile_config(
    cfg: EasyDict,
    env_manager=None,
    policy=None,
    learner=None,
    collector=None,
    evaluator=None,
    buffer=None,
    env=None,
    reward_model=None,
    world_model=None,
    seed: int = 0,
    auto: bool = False,
    create_cfg: Optional[EasyDict] = None,
    save_cfg: bool = True,
    save_path: str = 'total_config.py',
    renew_dir: bool = True,
) -> EasyDict:
    """Merge defaults of every component with the user config; returns the
    total config used everywhere downstream."""
    cfg = EasyDict(deepcopy(cfg))
    if create_cfg is not None and len(create_cfg) > 0:
        # resolve types from create_cfg
        if policy is None and 'policy' in create_cfg:
            from ding.policy import get_policy_cls
            policy = get_policy_cls(create_cfg.policy)
            cfg.policy = deep_merge_dicts({'type': create_cfg.policy.type}, cfg.get('policy', {}))
            cfg.policy.type = create_cfg.policy.type
            if 'import_names' in create_cfg.policy:
                cfg.policy.import_names = create_cfg.policy.import_names
        if 'env' in create_cfg and 'type' in create_cfg.env:
            cfg.env = deep_merge_dicts({'type': create_cfg.env.type}, cfg.get('env', {}))
            if 'import_names' in create_cfg.env:
                cfg.env.import_names = create_cfg.env.import_names
        if 'env_manager' in create_cfg:
            cfg.env.manager = deep_merge_dicts(cfg.env.get('manager', {}), {'type': create_cfg.env_manager.type})

    base = EasyDict(deepcopy(main_cfg_template))
    # policy defaults
    if policy is not None:
        policy_default = policy.default_config() if hasattr(policy, 'default_config') else EasyDict({})
        base.policy = policy_default
    # env manager defaults
    if env_manager is not None and hasattr(env_manager, 'default_config'):
        base.env.manager = deep_merge_dicts(base.env['manager'], env_manager.default_config())

    cfg = deep_merge_dicts(base, cfg)
    cfg = EasyDict(cfg)
    cfg.seed = seed

    # evaluator defaults (stop_value / eval freq plumbing)
    if 'stop_value' not in cfg.env:
        cfg.env.stop_value = float('inf')
    if 'n_evaluator_episode' not in cfg.env:
        cfg.env.n_evaluator_episode = cfg.env.get('evaluator_env_num', 1)
    if 'eval' not in cfg.policy or cfg.policy.eval is None:
        cfg.policy.eval = EasyDict({})
    if 'evaluator' not in cfg.policy.eval:
        cfg.policy.eval.evaluator = EasyDict({})
    if 'eval_freq' not in cfg.policy.eval.evaluator:
        cfg.policy.eval.evaluator.eval_freq = 100
    cfg.policy.eval.evaluator.stop_value = cfg.env.stop_value
    cfg.policy.eval.evaluator.n_episode = cfg.env.n_evaluator_episode

    # experiment dir
    if 'exp_name' not in cfg:
        cfg.exp_name = 'default_experiment'
    if save_cfg:
        if os.path.exists(cfg.exp_name) and renew_dir:
            stamp = datetime.datetime.now().strftime('%y%m%d_%H%M%S')
            cfg.exp_name += f'_{stamp}'
        os.makedirs(cfg.exp_name, exist_ok=True)
        save_config_py(cfg, os.path.join(cfg.exp_name, 'formatted_total_config.py'))
    return cfg


def compile_config_parallel(cfg: EasyDict, create_cfg: EasyDict, system_cfg: Optional[EasyDict] = None,
                            seed: int = 0, **kwargs) -> EasyDict:
    """Parallel-mode variant: same merge + system section defaults."""
    cfg = compile_config(cfg, seed=seed, create_cfg=create_cfg, **kwargs)
    if system_cfg is not None:
        cfg.system = EasyDict(system_cfg)
    return cfg


class Config:
    """Dict+text config pair loaded from a python file (reference
    ding/config/config.py Config:27). ``file_to_dict`` executes the file and
    keeps every non-dunder module-level value in ``cfg_dict``."""

    def __init__(self, cfg_dict: Optional[dict] = None, cfg_text: Optional[str] = None,
                 filename: Optional[str] = None) -> None:
        if cfg_dict is None:
            cfg_dict = {}
        if not isinstance(cfg_dict, dict):
            raise TypeError(f"invalid type for cfg_dict: {type(cfg_dict)}")
        self._cfg_dict = cfg_dict
        if cfg_text:
            self._text = cfg_text
        elif filename:
            with open(filename) as f:
                self._text = f.read()
        else:
            self._text = ''
        self._filename = filename

    @staticmethod
    def file_to_dict(filename: str) -> 'Config':
        spec = importlib.util.spec_from_file_location('ding_user_config_full', filename)
        module = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(module)
        cfg_dict = {k: v for k, v in vars(module).items() if not k.startswith('__') and not callable(v)
                    and not isinstance(v, type(importlib))}
        return Config(cfg_dict, filename=filename)

    @property
    def cfg_dict(self) -> dict:
        return self._cfg_dict


def read_config_directly(path: str) -> dict:
    """Whole-module config dict, no main/create splitting."""
    assert path.endswith('.py'), f"invalid config file suffix: {path}"
    return Config.file_to_dict(path).cfg_dict


def read_config_with_system(path: str) -> Tuple[EasyDict, EasyDict, EasyDict]:
    """(main_config, create_config, system_config) triple for parallel mode."""
    cfg = read_config_directly(path)
    for key in ('main_config', 'create_config', 'system_config'):
        assert key in cfg, f"a '{key}' variable must be declared in {path}"
    return EasyDict(cfg['main_config']), EasyDict(cfg['create_config']), EasyDict(cfg['system_config'])


_DEFAULT_HOST, _DEFAULT_PORT = '127.0.0.1', 50515


def parallel_transform(cfg: dict, coordinator_host: Optional[str] = None, learner_host=None,
                       collector_host=None) -> EasyDict:
    """Fill the system section (hosts/ports for coordinator, learners and
    collectors) of a parallel-mode config (reference config/utils.py:195)."""
    cfg = EasyDict(cfg)
    coordinator_host = coordinator_host or _DEFAULT_HOST
    learner_host = learner_host or [_DEFAULT_HOST]
    collector_host = collector_host or [_DEFAULT_HOST]
    if isinstance(learner_host, str):
        learner_host = [learner_host]
    if isinstance(collector_host, str):
        collector_host = [collector_host]
    system = cfg.get('system', EasyDict({}))
    system.coordinator = EasyDict({'host': coordinator_host, 'port': _DEFAULT_PORT})
    system.learners = EasyDict({
        f'learner{i}': EasyDict({'host': h, 'port': _DEFAULT_PORT + 1 + i}) for i, h in enumerate(learner_host)
    })
    system.collectors = EasyDict({
        f'collector{i}': EasyDict({'host': h, 'port': _DEFAULT_PORT + 101 + i}) for i, h in enumerate(collector_host)
    })
    cfg.system = system
    return cfg


def parallel_transform_slurm(cfg: dict, coordinator_host: Optional[str] = None, learner_node=None,
                             collector_node=None) -> EasyDict:
    """Slurm variant: node names resolve to hosts via node_to_host."""
    from ding.utils import find_free_port_slurm
    from ding.utils.misc_helpers import node_to_host
    learner_host = [node_to_host(n) for n in (learner_node or [])] or None
    collector_host = [node_to_host(n) for n in (collector_node or [])] or None
    out = parallel_transform(cfg, coordinator_host, learner_host, collector_host)
    # slurm boxes share nodes: replace fixed ports with job-derived free ones
    for grp in (out.system.learners, out.system.collectors):
        for v in grp.values():
            v.port = find_free_port_slurm(v.host)
    return out
