"""More specialized serial pipelines: NGU, R2D3, preference-based IRL (TREX),
guided cost, TD3-VAE, on-policy PPG, BCO, procedure cloning, plus the TREX
demonstration-collection application entry.

Parity: reference ding/entry/serial_entry_ngu.py, serial_entry_r2d3.py,
serial_entry_preference_based_irl.py, serial_entry_preference_based_irl_onpolicy.py,
serial_entry_guided_cost.py, serial_entry_td3_vae.py,
serial_entry_onpolicy_ppg.py, serial_entry_bco.py, serial_entry_pc.py,
application_entry_trex_collect_data.py.
"""
import copy
from typing import Any, List, Optional, Tuple, Union

import torch

from ding.policy import create_policy
from ding.reward_model import create_reward_model
from ding.utils import EasyDict
from ding.worker import BaseLearner, BaseSerialCommander, InteractionSerialEvaluator, create_buffer, \
    create_serial_collector
from .serial_entry import _prepare, random_collect
from .serial_entry_variants import _build_workers


def serial_pipeline_ngu(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """NGU: R2D2-style learner + dual intrinsic reward (RND lifelong novelty x
    episodic novelty), both trained online from collected unrolls."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    device = policy._get_attribute('device')
    rnd_reward_model = create_reward_model(cfg.rnd_reward_model, device=device)
    episodic_reward_model = create_reward_model(cfg.episodic_reward_model, device=device)
    if cfg.policy.get('random_collect_size', 0) > 0:
        random_collect(cfg.policy, policy, collector, collector_env, commander, replay_buffer)
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        rnd_reward_model.collect_data(new_data)
        episodic_reward_model.collect_data(new_data)
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        rnd_reward_model.train()
        episodic_reward_model.train()
        for _ in range(cfg.policy.learn.update_per_collect):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            # intrinsic reward fusion: r_i = r_episodic * clip(alpha_rnd, 1, L)
            episodic_r = episodic_reward_model.estimate(train_data)
            alpha = rnd_reward_model.estimate(train_data)
            intrinsic = torch.as_tensor(episodic_r).flatten() * torch.clamp(torch.as_tensor(alpha).flatten(), 1.0, 5.0)
            nu = cfg.policy.get('nu', 0.01)
            for d, r_i in zip(train_data, intrinsic):
                r = d['reward']
                if isinstance(r, (list, tuple)):  # r2d2 unroll: spread over steps
                    d['reward'] = [torch.as_tensor(x, dtype=torch.float32) + nu * float(r_i) for x in r]
                elif isinstance(r, torch.Tensor):
                    d['reward'] = r + nu * r_i.to(r.dtype)
                else:
                    d['reward'] = float(r) + nu * float(r_i)
            learner.train(train_data, collector.envstep)
            if learner.policy.get_attribute('priority'):
                replay_buffer.update(learner.priority_info)
        rnd_reward_model.clear_data()
        episodic_reward_model.clear_data()
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def serial_pipeline_r2d3(
    input_cfg: Union[str, Tuple[dict, dict]],
    expert_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    expert_model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """R2D3: recurrent DQfD — an expert buffer filled once from an expert
    policy plus an agent buffer filled online; each minibatch mixes the two
    with ratio ``pho`` (expert fraction)."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    if isinstance(expert_cfg, str):
        from ding.config import read_config
        expert_cfg = read_config(expert_cfg)
    e_cfg, e_create_cfg = copy.deepcopy(expert_cfg)
    if not e_create_cfg.policy.type.endswith('_command'):
        e_create_cfg.policy.type = e_create_cfg.policy.type + '_command'
    from ding.config import compile_config
    e_cfg = compile_config(e_cfg, seed=seed, auto=True, create_cfg=e_create_cfg, save_cfg=False)
    expert_policy = create_policy(e_cfg.policy, model=expert_model, enable_field=['collect', 'command'])
    model_path = e_cfg.policy.collect.get('model_path', None)
    if model_path:
        expert_policy.collect_mode.load_state_dict(torch.load(model_path, map_location='cpu'))
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    expert_collector = create_serial_collector(
        e_cfg.policy.collect.get('collector', EasyDict({'type': 'sample'})), env=collector_env,
        policy=expert_policy.collect_mode, exp_name=cfg.exp_name
    )
    expert_buffer = create_buffer(e_cfg.policy.other.replay_buffer, exp_name=cfg.exp_name)
    learner.call_hook('before_run')
    # one-time expert fill
    expert_size = e_cfg.policy.learn.get('expert_replay_buffer_size', 1000)
    if expert_size > 0:
        expert_data = expert_collector.collect(n_sample=expert_size, train_iter=0, policy_kwargs={'eps': -1})
        for d in expert_data:
            d['is_expert'] = 1
        expert_buffer.push(expert_data, cur_collector_envstep=0)
    collector.reset(policy.collect_mode)
    pho = cfg.policy.collect.get('pho', 0.25)
    batch_size = learner.policy.get_attribute('batch_size')
    expert_bs = int(batch_size * pho)
    agent_bs = batch_size - expert_bs
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
        for _ in range(cfg.policy.learn.update_per_collect):
            agent_part = replay_buffer.sample(agent_bs, learner.train_iter)
            expert_part = expert_buffer.sample(expert_bs, learner.train_iter) if expert_bs > 0 else []
            if agent_part is None:
                break
            train_data = list(agent_part) + list(expert_part or [])
            learner.train(train_data, collector.envstep)
            if learner.policy.get_attribute('priority'):
                replay_buffer.update(learner.priority_info)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def serial_pipeline_preference_based_irl(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """TREX off-policy: pretrain a trajectory-ranking reward model once, then
    run standard off-policy RL on the learned reward."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    reward_model = create_reward_model(cfg.reward_model, device=policy._get_attribute('device'))
    reward_model.train()  # one-shot pretraining from ranked demos
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        for _ in range(cfg.policy.learn.update_per_collect):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            train_data = reward_model.estimate(train_data)
            learner.train(train_data, collector.envstep)
            if learner.policy.get_attribute('priority'):
                replay_buffer.update(learner.priority_info)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def serial_pipeline_preference_based_irl_onpolicy(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """TREX on-policy: pretrained ranking reward + on-policy learner (PPO)."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    reward_model = create_reward_model(cfg.reward_model, device=policy._get_attribute('device'))
    reward_model.train()
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        new_data = reward_model.estimate(new_data)
        learner.train(new_data, collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


# alias used in some reference examples
serial_pipeline_trex = serial_pipeline_preference_based_irl
serial_pipeline_trex_onpolicy = serial_pipeline_preference_based_irl_onpolicy


def serial_pipeline_guided_cost(
    input_cfg: Union[str, Tuple[dict, dict]],
    expert_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    expert_model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """GCL: alternate cost (reward) model updates on expert-vs-agent samples
    with policy updates on the learned cost."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    if isinstance(expert_cfg, str):
        from ding.config import read_config
        expert_cfg = read_config(expert_cfg)
    e_cfg, e_create_cfg = copy.deepcopy(expert_cfg)
    from ding.config import compile_config
    if not e_create_cfg.policy.type.endswith('_command'):
        e_create_cfg.policy.type = e_create_cfg.policy.type + '_command'
    e_cfg = compile_config(e_cfg, seed=seed, auto=True, create_cfg=e_create_cfg, save_cfg=False)
    expert_policy = create_policy(e_cfg.policy, model=expert_model, enable_field=['collect', 'command'])
    model_path = e_cfg.policy.collect.get('model_path', None)
    if model_path:
        expert_policy.collect_mode.load_state_dict(torch.load(model_path, map_location='cpu'))
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    expert_collector = create_serial_collector(
        e_cfg.policy.collect.get('collector', EasyDict({'type': 'sample'})), env=collector_env,
        policy=expert_policy.collect_mode, exp_name=cfg.exp_name
    )
    expert_buffer = create_buffer(e_cfg.policy.other.replay_buffer, exp_name=cfg.exp_name)
    reward_model = create_reward_model(cfg.reward_model, device=policy._get_attribute('device'))
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        expert_data = expert_collector.collect(train_iter=learner.train_iter, policy_kwargs={'eps': -1})
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        expert_buffer.push(expert_data, cur_collector_envstep=collector.envstep)
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        for _ in range(cfg.reward_model.get('update_per_collect', 1)):
            e_batch = expert_buffer.sample(cfg.reward_model.batch_size, learner.train_iter)
            a_batch = replay_buffer.sample(cfg.reward_model.batch_size, learner.train_iter)
            if e_batch is None or a_batch is None:
                break
            reward_model.train(e_batch, a_batch)
        for _ in range(cfg.policy.learn.update_per_collect):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            train_data = reward_model.estimate(train_data)
            learner.train(train_data, collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def serial_pipeline_td3_vae(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """HyAR/TD3-VAE: warm-up VAE phase on random data, then alternating
    RL-phase / VAE-phase minibatches flagged per sample."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    replay_buffer_recent = create_buffer(cfg.policy.other.replay_buffer, exp_name=cfg.exp_name + '_recent')
    if cfg.policy.get('random_collect_size', 0) > 0:
        random_collect(cfg.policy, policy, collector, collector_env, commander, replay_buffer)
        for _ in range(cfg.policy.learn.get('warm_up_update', 0)):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            for item in train_data:
                item['warm_up'] = True
            learner.train(train_data, collector.envstep)
        replay_buffer.clear()
        collector.reset(policy.collect_mode)
    learner.call_hook('before_run')
    count = 0
    rl_circle = cfg.policy.learn.get('rl_vae_update_circle', 1)
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        for item in new_data:
            item['warm_up'] = False
        replay_buffer.push(new_data, cur_collector_envstep=collector.envstep)
        replay_buffer_recent.push(copy.deepcopy(new_data), cur_collector_envstep=collector.envstep)
        # rl phase
        for _ in range(cfg.policy.learn.get('update_per_collect_rl', cfg.policy.learn.get('update_per_collect', 1))):
            train_data = replay_buffer.sample(learner.policy.get_attribute('batch_size'), learner.train_iter)
            if train_data is None:
                break
            for item in train_data:
                item['rl_phase'], item['vae_phase'] = True, False
            learner.train(train_data, collector.envstep)
        # vae phase on recent data
        if (count + 1) % rl_circle == 0:
            for _ in range(cfg.policy.learn.get('update_per_collect_vae', 0)):
                train_data = replay_buffer_recent.sample(
                    learner.policy.get_attribute('batch_size'), learner.train_iter
                )
                if train_data is None:
                    break
                for item in train_data:
                    item['rl_phase'], item['vae_phase'] = False, True
                learner.train(train_data, collector.envstep)
            replay_buffer_recent.clear()
        count += 1
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def serial_pipeline_onpolicy_ppg(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """PPG on-policy: collect -> learner.train directly (the policy handles
    the aux-phase scheduling internally)."""
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        learner.train(new_data, collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def serial_pipeline_bco(
    input_cfg: Union[str, Tuple[dict, dict]],
    expert_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    expert_model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """BCO(alpha): learn an inverse-dynamics model from self-play transitions,
    label expert observation pairs with inferred actions, behavior-clone."""
    import torch.nn as nn
    import torch.nn.functional as F
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    e_cfg, e_create_cfg = copy.deepcopy(expert_cfg)
    from ding.config import compile_config
    if not e_create_cfg.policy.type.endswith('_command'):
        e_create_cfg.policy.type = e_create_cfg.policy.type + '_command'
    e_cfg = compile_config(e_cfg, seed=seed, auto=True, create_cfg=e_create_cfg, save_cfg=False)
    expert_policy = create_policy(e_cfg.policy, model=expert_model, enable_field=['collect', 'command'])
    model_path = e_cfg.policy.collect.get('model_path', None)
    if model_path:
        expert_policy.collect_mode.load_state_dict(torch.load(model_path, map_location='cpu'))
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    expert_collector = create_serial_collector(
        e_cfg.policy.collect.get('collector', EasyDict({'type': 'sample'})), env=collector_env,
        policy=expert_policy.collect_mode, exp_name=cfg.exp_name
    )
    # expert demo: observation pairs only (BCO assumption: no expert actions)
    expert_data = expert_collector.collect(
        n_sample=cfg.policy.collect.get('n_sample', 1000), train_iter=0, policy_kwargs={'eps': -1}
    )
    obs_dim = expert_data[0]['obs'].numel()
    continuous = cfg.policy.get('continuous', False)
    act_dim = expert_data[0]['action'].numel() if continuous else int(cfg.policy.model.action_shape)
    idm = nn.Sequential(nn.Linear(2 * obs_dim, 128), nn.ReLU(), nn.Linear(128, act_dim))
    idm_opt = torch.optim.Adam(idm.parameters(), lr=1e-3)
    learner.call_hook('before_run')
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        # 1. self-play transitions train the inverse dynamics model
        new_data = collector.collect(train_iter=learner.train_iter, policy_kwargs=collect_kwargs)
        obs = torch.stack([torch.cat([d['obs'].flatten(), d['next_obs'].flatten()]) for d in new_data])
        act = torch.stack([d['action'] for d in new_data]).squeeze(-1)
        for _ in range(cfg.policy.learn.get('idm_train_epoch', 5)):
            pred = idm(obs)
            idm_loss = F.mse_loss(pred, act.float()) if continuous else F.cross_entropy(pred, act.long())
            idm_opt.zero_grad()
            idm_loss.backward()
            idm_opt.step()
        # 2. label expert pairs with inferred actions, then BC
        with torch.no_grad():
            e_obs = torch.stack([torch.cat([d['obs'].flatten(), d['next_obs'].flatten()]) for d in expert_data])
            inferred = idm(e_obs)
            inferred = inferred if continuous else inferred.argmax(dim=-1)
        bc_data = []
        for d, a in zip(expert_data, inferred):
            nd = dict(d)
            nd['action'] = a if continuous else a.unsqueeze(0)
            bc_data.append(nd)
        learner.train(bc_data, collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def serial_pipeline_pc(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    model: Optional[torch.nn.Module] = None,
    max_iter: int = int(1e6),
) -> Tuple['Policy', bool]:  # noqa
    """Procedure cloning: supervised training over a provided dataset of
    (obs, action-sequence) pairs; dataset construction is env-specific and
    supplied through ``cfg.policy.learn.train_epoch`` + a data loader in cfg."""
    from ding.config import compile_config, read_config
    if isinstance(input_cfg, str):
        cfg, create_cfg = read_config(input_cfg)
    else:
        cfg, create_cfg = copy.deepcopy(input_cfg)
    if not create_cfg.policy.type.endswith('_command'):
        create_cfg.policy.type = create_cfg.policy.type + '_command'
    cfg = compile_config(cfg, seed=seed, auto=True, create_cfg=create_cfg, save_cfg=True)
    policy = create_policy(cfg.policy, model=model, enable_field=['learn', 'eval'])
    learner = BaseLearner(cfg.policy.learn.get('learner', EasyDict({})), policy.learn_mode, exp_name=cfg.exp_name)
    dataset_fn = cfg.policy.learn.get('dataset_fn', None)
    assert dataset_fn is not None, "serial_pipeline_pc requires cfg.policy.learn.dataset_fn -> iterable of batches"
    learner.call_hook('before_run')
    stop = False
    for epoch in range(cfg.policy.learn.get('train_epoch', 1)):
        for batch in dataset_fn():
            learner.train(batch)
            if learner.train_iter >= max_iter:
                stop = True
                break
        if stop:
            break
    learner.call_hook('after_run')
    return policy, stop


def trex_collecting_data(args=None) -> None:
    """Collect ranked demonstration episodes for TREX from checkpoints of a
    partially-trained policy at several training stages.

    Parity: reference ding/entry/application_entry_trex_collect_data.py.
    Writes episode observations + returns into ``cfg.reward_model.data_path``
    via ``collect_episodic_demo_data``.
    """
    import os
    import pickle
    from .application_entry import collect_episodic_demo_data
    if isinstance(args, (tuple, list)):
        cfg, create_cfg = args
    else:
        cfg, create_cfg = args.cfg if hasattr(args, 'cfg') else args, None
    exp_path = cfg.reward_model.get('expert_model_path', cfg.exp_name)
    ckpts = sorted(
        f for f in (os.listdir(os.path.join(exp_path, 'ckpt')) if os.path.isdir(os.path.join(exp_path, 'ckpt')) else [])
        if f.endswith('.pth.tar')
    )
    data_path = cfg.reward_model.data_path
    os.makedirs(data_path, exist_ok=True)
    episodes, returns = [], []
    for i, ck in enumerate(ckpts):
        out_file = os.path.join(data_path, f'demo_{i}.pkl')
        collect_episodic_demo_data(
            (copy.deepcopy(cfg), copy.deepcopy(create_cfg)), seed=0, collect_count=1,
            expert_data_path=out_file, state_dict_path=os.path.join(exp_path, 'ckpt', ck)
        )
        with open(out_file, 'rb') as f:
            ep = pickle.load(f)
        episodes.append([t['obs'] for t in ep[0]])
        returns.append(sum(float(t['reward']) for t in ep[0]))
    with open(os.path.join(data_path, 'episodes_data.pkl'), 'wb') as f:
        pickle.dump(episodes, f)
    with open(os.path.join(data_path, 'learning_returns.pkl'), 'wb') as f:
        pickle.dump(returns, f)


def serial_pipeline_dreamer(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """DreamerV3: RSSM world model trained on sequence batches; policy trained
    purely in latent imagination from the posterior; collect/eval thread the
    RSSM filter state (reference serial_entry_mbrl.py:249)."""
    from ding.world_model import create_world_model
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, env_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    world_model = create_world_model(cfg.world_model, env=None)
    learner.call_hook('before_run')
    from .serial_entry import random_collect as _random_collect
    if cfg.policy.get('random_collect_size', 0) > 0:
        _random_collect(cfg.policy, policy, collector, collector_env, commander, env_buffer)
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(
                learner.save_checkpoint, learner.train_iter, collector.envstep,
                policy_kwargs=dict(world_model=world_model)
            )
            if stop:
                break
        steps = cfg.world_model.get('pretrain', 1) if world_model.should_pretrain() \
            else int(world_model.should_train(collector.envstep))
        for _ in range(steps):
            post, context = world_model.train(
                env_buffer, collector.envstep, learner.train_iter,
                learner.policy.get_attribute('batch_size'), cfg.policy.learn.batch_length
            )
            learner.train(
                post, collector.envstep,
                policy_kwargs=dict(world_model=world_model, envstep=collector.envstep)
            )
        data = collector.collect(
            train_iter=learner.train_iter,
            policy_kwargs=dict(world_model=world_model, envstep=collector.envstep)
        )
        env_buffer.push(data, cur_collector_envstep=collector.envstep)
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy


def generate_seeds(num_seeds: int = 500, base: int = 0) -> list:
    """Training level seed universe for PLR."""
    return list(range(base, base + num_seeds))


def serial_pipeline_plr(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
    max_env_step: int = int(1e10),
) -> 'Policy':  # noqa
    """Prioritized Level Replay: on-policy training where each collect cycle
    reseeds collector envs from the PLR level sampler (reference
    serial_entry_plr.py)."""
    from ding.data import LevelSampler
    from ding.policy.common_utils import default_preprocess_learn
    cfg, policy, collector_env, evaluator_env = _prepare(input_cfg, seed, env_setting, model)
    learner, collector, evaluator, replay_buffer, commander = _build_workers(
        cfg, policy, collector_env, evaluator_env
    )
    env_num = collector_env.env_num
    train_seeds = generate_seeds(cfg.get('level_replay', {}).get('num_seeds', 128))
    level_sampler = LevelSampler(
        train_seeds, cfg.policy.model.obs_shape, cfg.policy.model.action_shape, env_num,
        cfg.get('level_replay', EasyDict({}))
    )
    learner.call_hook('before_run')
    seeds = [int(level_sampler.sample('sequential')) for _ in range(env_num)]
    level_seeds = torch.Tensor(seeds)
    collector_env.seed(seeds)
    collector_env.reset()
    while True:
        collect_kwargs = commander.step()
        if evaluator.should_eval(learner.train_iter):
            stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter, collector.envstep)
            if stop:
                break
        new_data = collector.collect(
            train_iter=learner.train_iter, level_seeds=level_seeds, policy_kwargs=collect_kwargs
        )
        learner.train(new_data, collector.envstep)
        stacked = default_preprocess_learn(new_data, ignore_done=cfg.policy.learn.ignore_done, use_nstep=False)
        level_sampler.update_with_rollouts(stacked, env_num)
        seeds = [int(level_sampler.sample()) for _ in range(env_num)]
        level_seeds = torch.Tensor(seeds)
        collector_env.seed(seeds)
        collector_env.reset()
        if collector.envstep >= max_env_step or learner.train_iter >= max_train_iter:
            break
    learner.call_hook('after_run')
    return policy
