"""Specialized serial pipelines: reward-model (IRL/exploration), SQIL, GAIL,
DQFD, TREX, BC, MBRL (Dyna/Dream).

Parity: reference ding/entry/serial_entry_reward_model.py, _sqil.py,
_gail.py, _dqfd.py, _trex.py, _bc.py, _mbrl.py.
"""
import copy
from functools import partial
from typing import Any, List, Optional, Tuple, Union

import torch

from ding.config import compile_config, read_config
from ding.envs import create_env_manager, get_vec_env_setting
from ding.policy import create_policy
from ding.reward_model import create_reward_model
from ding.utils import EasyDict, set_pkg_seed
from ding.worker import (
    BaseLearner, BaseSerialCommander, InteractionSerialEvaluator, create_buffer, create_serial_collector,
)
from .serial_entry import _prepare


def _build_workers(cfg, policy, collector_env, evaluator_env):
    learner = BaseLearner(cfg.policy.learn.get('learner', EasyDict({})), policy.learn_mode, exp_name=cfg.exp_name)
    collector = create_serial_collector(
        cfg.policy.collect.get('collector', EasyDict({'type': 'sample'})), env=collector_env,
        policy=policy.collect_mode, exp_name=cfg.exp_name
    )
    evaluator = InteractionSerialEvaluator(
        cfg.policy.eval.evaluator, evaluator_env, policy.eval_mode, exp_name=cfg.exp_name
    )
    rb_cfg = cfg.policy.get('other', EasyDict({})).get('replay_buffer', None)
    # on-policy cfgs have no buffer section; variants that need one get None
    replay_buffer = create_buffer(rb_cfg, exp_name=cfg.exp_name) if rb_cfg is not None else None
    commander = BaseSerialCommander(EasyDict({}), learner, collector, evaluator, replay_buffer, policy.command_mode)
    return learner, collector, evaluator, replay_buffer, commander


def serial_pipeline_reward_model(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
    cooptrain_reward: bool = True,
    pretrain_reward: bool = False,
) -> 'Policy':  # noqa
    """Online RL with a learned reward (RND/ICM/GAIL/...)."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    reward_model = create_reward_model(cfg.reward_model, device=policy._get_attribute('device'))
    if pretrain_reward:
        reward_model.train()
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        if cooptrain_reward:
            reward_model.collect_data(new_data)
            reward_model.train()
        if cfg.policy.on_policy:
            # on-policy lane (e.g. RND + onpolicy PPO): relabel and train on
            # the whole fresh batch, no buffer (reference serial_entry_onpolicy
            # + reward-model estimate)
            train_data = reward_model.estimate(new_data)
            learner.train(train_data, collector.envstep)
        else:
            replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
            for i in range(cfg.policy.learn.update_per_collect):
                train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
                if train_data is None:
                    break
                train_data = reward_model.estimate(train_data)
                learner.train(train_data, collector.envstep)
                if learner.policy.get_attribute('priority'):
                    replay_buffer.update(learner.priority_info)
        if cooptrain_reward:
            reward_model.clear_data()
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    collector.close()
    evaluator.close()
    learner.close()
    return policy


def serial_pipeline_sqil(
    input_cfg: Union[str, Tuple[dict, dict]],
    expert_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    expert_model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """SQIL: half expert transitions (reward 1), half agent (reward 0)."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    expert_cfg_c, expert_policy, expert_collector_env, _ = _prepare(expert_cfg, seed, None, expert_model)
    if expert_cfg_c.policy.get('collect', {}).get('model_path', None):
        state = torch.load(expert_cfg_c.policy.collect.model_path, map_location='cpu', weights_only=False)
        expert_policy.collect_mode.load_state_dict(state)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    from ding.worker import SampleSerialCollector
    expert_collector = SampleSerialCollector(
        EasyDict({}), expert_collector_env, expert_policy.collect_mode, exp_name=cfg.exp_name,
        instance_name='expert_collector'
    )
    expert_buffer = create_buffer(cfg.policy.other.replay_buffer, exp_name=cfg.exp_name, instance_name='expert_buffer')
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        for d in new_data:
            d['reward'] = torch.zeros_like(torch.as_tensor(d['reward'], dtype=torch.float32))
        exp_data = expert_collector.collect(train_iter=learner.train_iter, policy_kwargs={'eps': -1})
        for d in exp_data:
            d['reward'] = torch.ones_like(torch.as_tensor(d['reward'], dtype=torch.float32))
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        expert_buffer.push(exp_data, cur_collector_envstep=collector.envstep)
        for i in range(cfg.policy.learn.update_per_collect):
            bs = learner.policy.get_attribute('batch_size')
            agent_batch = replay_buffer.sample(bs // 2, learner.train_iter)
            expert_batch = expert_buffer.sample(bs - bs // 2, learner.train_iter)
            if agent_batch is None or expert_batch is None:
                break
            learner.train(list(agent_batch) + list(expert_batch), collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    collector.close()
    expert_collector.close()
    evaluator.close()
    learner.close()
    return policy


def serial_pipeline_gail(
    input_cfg: Union[str, Tuple[dict, dict]],
    expert_data: Union[str, list],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """GAIL: discriminator-shaped rewards from expert demonstrations."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    reward_model = create_reward_model(cfg.reward_model, device=policy._get_attribute('device'))
    if isinstance(expert_data, str):
        import pickle
        with open(expert_data, 'rb') as f:
            expert_data = pickle.load(f)
    reward_model.load_expert_data(expert_data)
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        reward_model.collect_data(new_data)
        reward_model.train()
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        for i in range(cfg.policy.learn.update_per_collect):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            train_data = reward_model.estimate(train_data)
            learner.train(train_data, collector.envstep)
        reward_model.clear_data()
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    collector.close()
    evaluator.close()
    learner.close()
    return policy


def serial_pipeline_dqfd(
    input_cfg: Union[str, Tuple[dict, dict]],
    expert_data: Union[str, list],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """DQfD: pretrain on expert transitions, then mix expert/agent batches."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    expert_buffer = create_buffer(cfg.policy.other.replay_buffer, exp_name=cfg.exp_name, instance_name='expert_buffer')
    if isinstance(expert_data, str):
        import pickle
        with open(expert_data, 'rb') as f:
            expert_data = pickle.load(f)
    for d in expert_data:
        d['is_expert'] = 1
        expert_buffer.push(d)
    learner.call_hook('before_run')
    # expert pretrain phase
    for _ in range(cfg.policy.get('pretrain_iterations', 0)):
        batch = expert_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
        if batch:
            learner.train(batch)
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        for d in new_data:
            d['is_expert'] = 0
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        for i in range(cfg.policy.learn.update_per_collect):
            bs = learner.policy.get_attribute('batch_size')
            agent_batch = replay_buffer.sample(bs // 2, learner.train_iter)
            expert_batch = expert_buffer.sample(bs - bs // 2, learner.train_iter)
            if agent_batch is None or expert_batch is None:
                break
            learner.train(list(agent_batch) + list(expert_batch), collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    collector.close()
    evaluator.close()
    learner.close()
    return policy


def serial_pipeline_bc(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int,
    data,
    max_iter=int(1e6),
) -> Tuple['Policy', bool]:  # noqa
    """Supervised behaviour cloning over a fixed transition set."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, None, None)
    learner = BaseLearner(cfg.policy.learn.get('learner', EasyDict({})), policy.learn_mode, exp_name=cfg.exp_name)
    evaluator = InteractionSerialEvaluator(
        cfg.policy.eval.evaluator, evaluator_env, policy.eval_mode, exp_name=cfg.exp_name
    )
    learner.call_hook('before_run')
    import random
    stop = False
    bs = cfg.policy.learn.batch_size
    for it in range(max_iter):
        batch = random.sample(data, min(bs, len(data)))
        learner.train(batch)
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter)
            if stop:
                break
    learner.call_hook('after_run')
    evaluator.close()
    learner.close()
    return policy, stop


def serial_pipeline_dyna(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """Dyna-style MBRL (MBPO): policy trained on real+imagined mixture."""
    from ding.world_model import create_world_model
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    world_model = create_world_model(cfg.world_model, env=None)
    img_buffer = create_buffer(
        cfg.world_model.other.imagination_buffer, exp_name=cfg.exp_name, instance_name='img_buffer'
    )
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        if world_model.should_train(collector.envstep):
            world_model.train(replay_buffer, collector.envstep, learner.train_iter)
            world_model.fill_img_buffer(
                policy.collect_mode, replay_buffer, img_buffer, collector.envstep, learner.train_iter
            )
        for i in range(cfg.policy.learn.update_per_collect):
            batch = world_model.sample(
                replay_buffer, img_buffer, learner.policy.get_attribute('batch_size'), learner.train_iter
            )
            if not batch:
                break
            learner.train(batch, collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    collector.close()
    evaluator.close()
    learner.close()
    return policy


def serial_pipeline_dream(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """Dream-style MBRL (MBSAC/STEVESAC): the policy trains on differentiable
    imagined rollouts, so the learner forwards the world model + envstep into
    policy._forward_learn (reference serial_entry_mbrl.py:176)."""
    from ding.world_model import create_world_model
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    world_model = create_world_model(cfg.world_model, env=None)
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        if world_model.should_train(collector.envstep):
            world_model.train(replay_buffer, collector.envstep, learner.train_iter)
        upc = max(1, cfg.policy.learn.update_per_collect // max(1, world_model.rollout_length_scheduler(collector.envstep)))
        for _ in range(upc):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            learner.train(
                train_data, collector.envstep,
                policy_kwargs=dict(world_model=world_model, envstep=collector.envstep)
            )
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy
