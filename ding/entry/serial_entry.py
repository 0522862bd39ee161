"""Legacy serial training pipeline.

Parity: reference ding/entry/serial_entry.py:18 (serial_pipeline) — the
SURVEY §3.1 call stack: env managers -> policy -> BaseLearner/collector/
evaluator/buffer/commander -> outer loop.
"""
import os
from functools import partial
from typing import Any, List, Optional, Tuple, Union

import torch

from ding.config import compile_config, read_config, save_config
from ding.envs import create_env_manager, get_vec_env_setting
from ding.policy import create_policy
from ding.utils import EasyDict, set_pkg_seed
from ding.worker import (
    BaseLearner, BaseSerialCommander, InteractionSerialEvaluator, SampleSerialCollector, EpisodeSerialCollector,
    create_buffer, create_serial_collector, create_serial_evaluator,
)


def _prepare(input_cfg, seed, env_setting, model, command: bool = True):
    if isinstance(input_cfg, str):
        cfg, create_cfg = read_config(input_cfg)
    else:
        cfg, create_cfg = input_cfg
        import copy as _copy
        cfg, create_cfg = _copy.deepcopy(cfg), _copy.deepcopy(create_cfg)
    if command and not create_cfg.policy.type.endswith('_command'):
        create_cfg.policy.type = create_cfg.policy.type + '_command'
    cfg = compile_config(cfg, seed=seed, auto=True, create_cfg=create_cfg, save_cfg=True)
    # env
    if env_setting is None:
        env_fn, collector_env_cfg, evaluator_env_cfg = get_vec_env_setting(cfg.env)
    else:
        env_fn, collector_env_cfg, evaluator_env_cfg = env_setting
    manager_cfg = EasyDict(dict(cfg.env.manager))
    manager_cfg.type = cfg.env.manager.get('type', 'base')
    collector_env = create_env_manager(manager_cfg, [partial(env_fn, cfg=c) for c in collector_env_cfg])
    evaluator_env = create_env_manager(manager_cfg, [partial(env_fn, cfg=c) for c in evaluator_env_cfg])
    collector_env.seed(cfg.seed)
    evaluator_env.seed(cfg.seed, dynamic_seed=False)
    set_pkg_seed(cfg.seed, use_cuda=cfg.policy.cuda)
    policy = create_policy(cfg.policy, model=model)
    return cfg, policy, collector_env, evaluator_env


def random_collect(policy_cfg, policy, collector, collector_env, commander, replay_buffer) -> None:
    assert policy_cfg.random_collect_size > 0
    collect_kwargs = commander.step()
    if policy_cfg.get('action_space', None) == 'continuous' or True:
        new_data = collector.collect(
            n_sample=policy_cfg.random_collect_size, random_collect=True, record_random_collect=False,
            policy_kwargs=collect_kwargs
        )
    replay_buffer.push(new_data, cur_collector_envstep=0)
    collector.reset_policy(policy.collect_mode)


def serial_pipeline(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
    dynamic_seed: Optional[bool] = None,
) -> 'Policy':  # noqa
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)

    tb_logger = None
    learner = BaseLearner(cfg.policy.learn.get('learner', EasyDict({})), policy.learn_mode, tb_logger,
                          exp_name=cfg.exp_name)
    collector = create_serial_collector(
        cfg.policy.collect.get('collector', EasyDict({'type': 'sample'})), env=collector_env,
        policy=policy.collect_mode, tb_logger=tb_logger, exp_name=cfg.exp_name
    )
    evaluator = InteractionSerialEvaluator(
        cfg.policy.eval.evaluator, evaluator_env, policy.eval_mode, tb_logger, exp_name=cfg.exp_name
    )
    replay_buffer = create_buffer(cfg.policy.other.replay_buffer, tb_logger=tb_logger, exp_name=cfg.exp_name)
    commander = BaseSerialCommander(
        EasyDict({}), learner, collector, evaluator, replay_buffer, policy.command_mode
    )

    learner.call_hook('before_run')
    if cfg.policy.get('random_collect_size', 0) > 0:
        random_collect(cfg.policy, policy, collector, collector_env, commander, replay_buffer)

    stop = False
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, eval_info = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        for i in range(cfg.policy.learn.update_per_collect):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            learner.train(train_data, collector.envstep)
            if learner.policy.get_attribute('priority'):
                replay_buffer.update(learner.priority_info)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break

    learner.call_hook('after_run')
    collector.close()
    evaluator.close()
    learner.close()
    return policy


def serial_pipeline_onpolicy(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """On-policy variant (PPO/A2C): no replay buffer, train on each collect."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    tb_logger = None
    learner = BaseLearner(cfg.policy.learn.get('learner', EasyDict({})), policy.learn_mode, tb_logger,
                          exp_name=cfg.exp_name)
    collector = create_serial_collector(
        cfg.policy.collect.get('collector', EasyDict({'type': 'sample'})), env=collector_env,
        policy=policy.collect_mode, tb_logger=tb_logger, exp_name=cfg.exp_name
    )
    evaluator = InteractionSerialEvaluator(
        cfg.policy.eval.evaluator, evaluator_env, policy.eval_mode, tb_logger, exp_name=cfg.exp_name
    )
    learner.call_hook('before_run')
    while True:
        if evaluator.should_eval(learner.train_iter):
            stop, eval_info = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter)
        if new_data and isinstance(new_data[0], list):
            # episode collector with get_train_sample: flatten episodes
            new_data = [t for episode in new_data for t in episode]
        learner.train(new_data, collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    collector.close()
    evaluator.close()
    learner.close()
    return policy
