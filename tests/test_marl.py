"""MARL smoke tests (QMIX/WQMIX/COMA/MAPPO on the cooperative matrix env)."""
import pytest
from ding.entry import serial_pipeline, serial_pipeline_onpolicy
from ding.utils import EasyDict


def _marl_cfg(ptype, model, extra=None):
    main = EasyDict(dict(
        exp_name=f'exp/test_{ptype}',
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=100),
        policy=dict(
            cuda=False,
            model=model,
            learn=dict(update_per_collect=1, batch_size=4, learning_rate=5e-4),
            collect=dict(n_sample=8, unroll_len=5, env_num=2),
            eval=dict(env_num=2, evaluator=dict(eval_freq=int(1e6))),
            other=dict(eps=dict(type='exp', start=1, end=0.05, decay=10000),
                       replay_buffer=dict(replay_buffer_size=200)),
        ),
    ))
    if extra:
        from ding.utils import deep_merge_dicts
        main.policy = EasyDict(deep_merge_dicts(main.policy, extra))
    create = EasyDict(dict(
        env=dict(type='coop_matrix', import_names=['dizoo.multiagent.envs.coop_matrix_env']),
        env_manager=dict(type='base'),
        policy=dict(type=ptype),
    ))
    return main, create


QMIX_MODEL = dict(agent_num=3, obs_shape=8, global_obs_shape=12, action_shape=4, hidden_size_list=[32, 32],
                  mixer=True)


def test_qmix():
    serial_pipeline(_marl_cfg('qmix', QMIX_MODEL), seed=0, max_train_iter=2)


def test_wqmix():
    serial_pipeline(_marl_cfg('wqmix', QMIX_MODEL), seed=0, max_train_iter=2)


def test_coma():
    model = dict(agent_num=3, obs_shape=dict(agent_state=8, global_state=12), action_shape=4)
    serial_pipeline(_marl_cfg('coma', model), seed=0, max_train_iter=2)


def test_mappo():
    model = dict(agent_obs_shape=8, global_obs_shape=12, action_shape=4, agent_num=3,
                 actor_hidden_size_list=[32, 32], critic_hidden_size_list=[32, 32])
    extra = dict(multi_agent=True, action_space='discrete',
                 learn=dict(epoch_per_collect=2, batch_size=8, learning_rate=3e-4),
                 collect=dict(n_sample=32, unroll_len=1, discount_factor=0.99, gae_lambda=0.95))
    main, create = _marl_cfg('ppo', model, extra)
    serial_pipeline_onpolicy((main, create), seed=0, max_train_iter=2)


def test_happo():
    extra = dict(
        agent_num=3,
        action_space='discrete',
        learn=dict(epoch_per_collect=1, batch_size=16, learning_rate=3e-4),
        collect=dict(n_sample=32, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
    )
    model = dict(agent_obs_shape=8, global_obs_shape=12, action_shape=4, agent_num=3,
                 actor_hidden_size_list=[32, 32], critic_hidden_size_list=[32, 32])
    main, create = _marl_cfg('happo', model, extra)
    serial_pipeline_onpolicy((main, create), seed=0, max_train_iter=2)


def test_atoc_policy():
    """ATOC: communicating MADDPG — collect with delta_q, learn all three losses."""
    import torch
    from ding.policy import create_policy
    from ding.policy.atoc import ATOCPolicy
    from ding.utils import EasyDict, deep_merge_dicts
    A, OBS, ACT = 3, 6, 2
    cfg = EasyDict(deep_merge_dicts(ATOCPolicy.default_config(), EasyDict(dict(
        type='atoc', cuda=False,
        model=dict(type='atoc', import_names=['ding.model.template.atoc'],
                   obs_shape=OBS, action_shape=ACT, thought_size=8, n_agent=A,
                   communication=True, agent_per_group=2),
        learn=dict(batch_size=8),
        collect=dict(n_sample=8, unroll_len=1),
    ))))
    pol = create_policy(cfg, enable_field=['learn', 'collect', 'eval'])
    out = pol._forward_collect({0: torch.randn(A, OBS), 1: torch.randn(A, OBS)})
    assert out[0]['action'].shape == (A, ACT)
    assert 'delta_q' in out[0]
    from ding.envs import BaseEnvTimestep
    trans = [
        pol._process_transition(
            torch.randn(A, OBS), out[0],
            BaseEnvTimestep(torch.randn(A, OBS), torch.randn(1), False, {})
        ) for _ in range(8)
    ]
    samples = pol._get_train_sample(trans)
    infos = pol._forward_learn(samples)
    assert 'critic_loss' in infos and 'actor_loss' in infos and 'attention_loss' in infos
    ev = pol._forward_eval({0: torch.randn(A, OBS)})
    assert ev[0]['action'].shape == (A, ACT)


def test_particle_spread_env():
    """MPE simple_spread stand-in: physics, shapes, episode accounting."""
    import numpy as np
    from dizoo.multiagent.envs.particle_env import ParticleSpreadEnv
    env = ParticleSpreadEnv({'agent_num': 3, 'max_step': 10})
    env.seed(0)
    obs = env.reset()
    assert obs['agent_state'].shape == (3, 14)
    assert obs['global_state'].shape == (18, )
    assert obs['action_mask'].shape == (3, 5)
    total = 0.0
    for t in range(10):
        ts = env.step(env.random_action())
        assert ts.obs['agent_state'].shape == (3, 14)
        assert ts.reward.shape == (1, ) and ts.reward[0] <= 0
        total += float(ts.reward[0])
        assert ts.done == (t == 9)
    assert abs(ts.info['eval_episode_return'] - total) < 1e-4
    # thrust must actually move the agents
    env.seed(1)
    env.reset()
    p0 = env._pos.copy()
    env.step(np.array([2, 2, 2]))  # +x thrust for everyone
    assert (env._pos[:, 0] > p0[:, 0]).all()


def test_qmix_on_particle_spread():
    model = dict(agent_num=3, obs_shape=14, global_obs_shape=18, action_shape=5,
                 hidden_size_list=[32, 32], mixer=True)
    main, create = _marl_cfg('qmix', model)
    main.exp_name = 'exp/test_qmix_particle'
    create.env = EasyDict(dict(type='particle_spread', import_names=['dizoo.multiagent.envs.particle_env']))
    serial_pipeline((main, create), seed=0, max_train_iter=2)
