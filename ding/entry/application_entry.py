"""Deploy-time entries: evaluation and demo-data collection.

Parity: reference ding/entry/application_entry.py (eval, collect_demo_data,
collect_episodic_demo_data, episode_to_transitions).
"""
import copy
import pickle
from functools import partial
from typing import Any, List, Optional, Tuple, Union

import torch

from ding.config import compile_config, read_config
from ding.envs import create_env_manager, get_vec_env_setting
from ding.policy import create_policy
from ding.utils import EasyDict, set_pkg_seed
from ding.utils.data import offline_data_save_type
from ding.worker import InteractionSerialEvaluator, EpisodeSerialCollector, SampleSerialCollector


def _setup(input_cfg, seed, env_setting, model, load_path=None, command=True):
    if isinstance(input_cfg, str):
        cfg, create_cfg = read_config(input_cfg)
    else:
        cfg, create_cfg = copy.deepcopy(input_cfg[0]), copy.deepcopy(input_cfg[1])
    if command and not create_cfg.policy.type.endswith('_command'):
        create_cfg.policy.type += '_command'
    cfg = compile_config(cfg, seed=seed, auto=True, create_cfg=create_cfg, save_cfg=False)
    if env_setting is None:
        env_fn, collector_env_cfg, evaluator_env_cfg = get_vec_env_setting(cfg.env)
    else:
        env_fn, collector_env_cfg, evaluator_env_cfg = env_setting
    manager_cfg = EasyDict(dict(cfg.env.manager))
    manager_cfg.type = cfg.env.manager.get('type', 'base')
    set_pkg_seed(seed, use_cuda=cfg.policy.cuda)
    policy = create_policy(cfg.policy, model=model)
    if load_path:
        state = torch.load(load_path, map_location='cpu', weights_only=False)
        policy.learn_mode.load_state_dict(state)
    return cfg, policy, env_fn, collector_env_cfg, evaluator_env_cfg, manager_cfg


def eval(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    state_dict: Optional[dict] = None,
    load_path: Optional[str] = None,
    replay_path: Optional[str] = None,
) -> float:
    cfg, policy, env_fn, _, evaluator_env_cfg, manager_cfg = _setup(input_cfg, seed, env_setting, model, load_path)
    evaluator_env = create_env_manager(manager_cfg, [partial(env_fn, cfg=c) for c in evaluator_env_cfg])
    evaluator_env.seed(seed, dynamic_seed=False)
    if state_dict is not None:
        policy.eval_mode.load_state_dict(state_dict)
    if replay_path:
        evaluator_env.enable_save_replay(replay_path)
    evaluator = InteractionSerialEvaluator(cfg.policy.eval.evaluator, evaluator_env, policy.eval_mode,
                                           exp_name=cfg.exp_name)
    _, episode_info = evaluator.eval()
    import numpy as np
    value = float(np.mean(episode_info['eval_episode_return']))
    evaluator.close()
    return value


def collect_demo_data(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int,
    collect_count: int,
    expert_data_path: Optional[str] = None,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    state_dict: Optional[dict] = None,
    state_dict_path: Optional[str] = None,
) -> List[dict]:
    cfg, policy, env_fn, collector_env_cfg, _, manager_cfg = _setup(input_cfg, seed, env_setting, model)
    collector_env = create_env_manager(manager_cfg, [partial(env_fn, cfg=c) for c in collector_env_cfg])
    collector_env.seed(seed)
    if state_dict is None and state_dict_path is not None:
        state_dict = torch.load(state_dict_path, map_location='cpu', weights_only=False)
    if state_dict is not None:
        policy.collect_mode.load_state_dict(state_dict)
    collector = SampleSerialCollector(
        EasyDict({}), collector_env, policy.collect_mode, exp_name=cfg.exp_name
    )
    policy_kwargs = {'eps': -1} if 'eps' in cfg.policy.other else {}
    exp_data = collector.collect(n_sample=collect_count, policy_kwargs=policy_kwargs or None)
    exp_data = [{k: v for k, v in d.items()} for d in exp_data]
    if expert_data_path:
        offline_data_save_type(
            exp_data, expert_data_path, data_type=cfg.policy.collect.get('data_type', 'naive')
        )
    collector.close()
    return exp_data


def collect_episodic_demo_data(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int,
    collect_count: int,
    expert_data_path: str,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    state_dict: Optional[dict] = None,
) -> List[list]:
    cfg, policy, env_fn, collector_env_cfg, _, manager_cfg = _setup(input_cfg, seed, env_setting, model)
    collector_env = create_env_manager(manager_cfg, [partial(env_fn, cfg=c) for c in collector_env_cfg])
    collector_env.seed(seed)
    if state_dict is not None:
        policy.collect_mode.load_state_dict(state_dict)
    collector = EpisodeSerialCollector(
        EpisodeSerialCollector.default_config(), collector_env, policy.collect_mode, exp_name=cfg.exp_name
    )
    policy_kwargs = {'eps': -1} if 'eps' in cfg.policy.other else {}
    episodes = collector.collect(n_episode=collect_count, policy_kwargs=policy_kwargs or None)
    with open(expert_data_path, 'wb') as f:
        pickle.dump(episodes, f)
    collector.close()
    return episodes


def episode_to_transitions(data_path: str, expert_data_path: str, nstep: int) -> None:
    """Flatten episodic demo data into n-step transitions."""
    from collections import deque
    from ding.rl_utils import get_nstep_return_data
    with open(data_path, 'rb') as f:
        episodes = pickle.load(f)
    out = []
    for ep in episodes:
        out.extend(list(get_nstep_return_data(deque(ep), nstep)))
    with open(expert_data_path, 'wb') as f:
        pickle.dump(out, f)


def episode_to_transitions_filter(data_path: str, expert_data_path: str, nstep: int, min_episode_return: float) -> None:
    """Like ``episode_to_transitions`` but drops episodes whose total return
    is below ``min_episode_return`` (reference application_entry.py:258)."""
    from collections import deque
    from ding.rl_utils import get_nstep_return_data
    with open(data_path, 'rb') as f:
        episodes = pickle.load(f)
    out = []
    for ep in episodes:
        ret = sum(float(step['reward'].sum() if hasattr(step['reward'], 'sum') else step['reward']) for step in ep)
        if ret < min_episode_return:
            continue
        out.extend(list(get_nstep_return_data(deque(ep), nstep)))
    with open(expert_data_path, 'wb') as f:
        pickle.dump(out, f)


def collect_episodic_demo_data_for_trex(
    input_cfg,
    seed: int,
    collect_count: int,
    rank: int,
    save_cfg_path: str,
    env_setting=None,
    model=None,
    state_dict=None,
    state_dict_path: str = None,
):
    """TREX demo collection from one checkpoint stage: collect episodes with
    the (partially trained) policy and write them under ``save_cfg_path``
    tagged by checkpoint rank (reference application_entry_trex_collect_data.py:18)."""
    import os
    if state_dict is None and state_dict_path is not None:
        state_dict = torch.load(state_dict_path, map_location='cpu')
    os.makedirs(save_cfg_path, exist_ok=True)
    expert_data_path = os.path.join(save_cfg_path, f'episodes_data_{rank}.pkl')
    return collect_episodic_demo_data(
        input_cfg, seed, collect_count, expert_data_path, env_setting=env_setting, model=model, state_dict=state_dict
    )
