"""Smoke-level training for every registered algorithm: 1-2 serial_pipeline
iterations on CartPole (discrete) / Pendulum (continuous), mirroring the
reference's entry/tests/test_serial_entry.py strategy."""
import copy

import pytest

from ding.entry import serial_pipeline, serial_pipeline_onpolicy
from ding.utils import EasyDict


def cartpole_cfg(policy_type: str, extra_policy: dict = None, buffer_type: str = 'naive') -> tuple:
    main = EasyDict(dict(
        exp_name=f'exp/test_{policy_type}',
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=195),
        policy=dict(
            cuda=False,
            model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[32, 32]),
            nstep=3,
            discount_factor=0.97,
            learn=dict(update_per_collect=2, batch_size=16, learning_rate=1e-3),
            collect=dict(n_sample=16),
            eval=dict(evaluator=dict(eval_freq=int(1e6))),
            other=dict(
                eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                replay_buffer=dict(type=buffer_type, replay_buffer_size=1000),
            ),
        ),
    ))
    if extra_policy:
        from ding.utils import deep_merge_dicts
        main.policy = EasyDict(deep_merge_dicts(main.policy, extra_policy))
    create = EasyDict(dict(
        env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
        env_manager=dict(type='base'),
        policy=dict(type=policy_type),
    ))
    return main, create


def pendulum_cfg(policy_type: str, extra_policy: dict = None) -> tuple:
    main = EasyDict(dict(
        exp_name=f'exp/test_{policy_type}',
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=-200, act_scale=True),
        policy=dict(
            cuda=False,
            random_collect_size=24,
            model=dict(obs_shape=3, action_shape=1),
            learn=dict(update_per_collect=2, batch_size=16),
            collect=dict(n_sample=16),
            eval=dict(evaluator=dict(eval_freq=int(1e6))),
            other=dict(replay_buffer=dict(replay_buffer_size=1000)),
        ),
    ))
    if extra_policy:
        from ding.utils import deep_merge_dicts
        main.policy = EasyDict(deep_merge_dicts(main.policy, extra_policy))
    create = EasyDict(dict(
        env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
        env_manager=dict(type='base'),
        policy=dict(type=policy_type),
    ))
    return main, create


@pytest.mark.parametrize('ptype', ['dqn', 'c51', 'qrdqn', 'iqn', 'rainbow', 'sql', 'mdqn', 'sqn'])
def test_value_based_smoke(ptype):
    extra = dict(nstep=1) if ptype in ('mdqn', 'sqn') else None
    main, create = cartpole_cfg(ptype, extra_policy=extra)
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_fqf_smoke():
    main, create = cartpole_cfg('fqf')
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_per_buffer_smoke():
    main, create = cartpole_cfg('dqn', extra_policy=dict(priority=True, priority_IS_weight=True),
                                buffer_type='advanced')
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_a2c_smoke():
    main, create = cartpole_cfg('a2c')
    main.policy.learn.pop('update_per_collect', None)
    serial_pipeline_onpolicy((main, create), seed=0, max_train_iter=2)


def test_impala_smoke():
    main, create = cartpole_cfg('impala', extra_policy=dict(unroll_len=8, learn=dict(batch_size=2)))
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_ppo_offpolicy_smoke():
    main, create = cartpole_cfg(
        'ppo_offpolicy',
        extra_policy=dict(model=dict(action_space='discrete'), learn=dict(epoch_per_collect=1))
    )
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_ddpg_smoke():
    main, create = pendulum_cfg(
        'ddpg', extra_policy=dict(model=dict(action_space='regression'),
                                  learn=dict(learning_rate_actor=1e-3, learning_rate_critic=1e-3))
    )
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_td3_smoke():
    main, create = pendulum_cfg(
        'td3', extra_policy=dict(model=dict(action_space='regression', twin_critic=True),
                                 learn=dict(learning_rate_actor=1e-3, learning_rate_critic=1e-3))
    )
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_sac_smoke():
    main, create = pendulum_cfg(
        'sac', extra_policy=dict(model=dict(action_space='reparameterization', twin_critic=True))
    )
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_discrete_sac_smoke():
    main, create = cartpole_cfg('discrete_sac', extra_policy=dict(model=dict(twin_critic=True)))
    main.policy.random_collect_size = 0
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_r2d2_smoke():
    main, create = cartpole_cfg(
        'r2d2',
        extra_policy=dict(
            priority=True, priority_IS_weight=True, nstep=2, burnin_step=2, learn_unroll_len=6,
            model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[32, 32]),
            learn=dict(update_per_collect=1, batch_size=4, learning_rate=1e-4),
            collect=dict(n_sample=8, env_num=2), eval=dict(env_num=2, evaluator=dict(eval_freq=int(1e6))),
        ),
        buffer_type='advanced',
    )
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_ppg_smoke():
    main, create = cartpole_cfg('ppg_offpolicy', extra_policy=dict(
        model=dict(action_space='discrete'),
        learn=dict(epoch_per_collect=1, aux_freq=1, aux_train_epoch=1, batch_size=16)))
    serial_pipeline((main, create), seed=0, max_train_iter=3)


def test_acer_smoke():
    main, create = cartpole_cfg('acer', extra_policy=dict(unroll_len=8, learn=dict(batch_size=2)))
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_dqfd_policy_smoke():
    main, create = cartpole_cfg('dqfd', extra_policy=dict(nstep=3))
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_d4pg_smoke():
    main, create = pendulum_cfg('d4pg', extra_policy=dict(
        nstep=2, model=dict(action_space='regression', v_min=-100, v_max=100, n_atom=51)))
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_ppof_policy():
    """PPOF simplified high-level PPO: collect -> train -> eval round trip."""
    import torch
    from ding.policy.ppof import PPOFPolicy
    from ding.model.template.vac import VAC
    cfg = PPOFPolicy.default_config()
    cfg.cuda = False
    cfg.batch_size = 8
    cfg.epoch_per_collect = 2
    cfg.n_sample = 32
    model = VAC(obs_shape=4, action_shape=2, encoder_hidden_size_list=[16, 16])
    pol = PPOFPolicy(cfg, model)
    out = pol.collect(torch.randn(32, 4))
    assert out['action'].shape == (32, )
    data = {
        'obs': torch.randn(32, 4), 'next_obs': torch.randn(32, 4),
        'action': out['action'], 'logit': out['logit'],
        'reward': torch.randn(32), 'done': torch.zeros(32),
    }
    infos = pol.forward(data)
    assert len(infos) == 2 * (32 // 8)
    assert all(abs(i['total_loss']) < 1e6 for i in infos)
    ev = pol.eval(torch.randn(5, 4))
    assert ev['action'].shape == (5, )
    sd = pol.state_dict()
    pol.load_state_dict(sd)


def test_minigrid_lite_rnd_pipeline():
    """Sparse-reward gridworld + RND intrinsic reward end-to-end."""
    from ding.entry import serial_pipeline_reward_model
    main = EasyDict(dict(
        exp_name='exp/test_minigrid_rnd',
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=2,
                 grid_size=5, max_step=30),
        policy=dict(
            cuda=False, nstep=1, discount_factor=0.99,
            model=dict(obs_shape=5 * 5 * 4 + 4, action_shape=3, encoder_hidden_size_list=[32, 32]),
            learn=dict(update_per_collect=2, batch_size=16, learning_rate=1e-3),
            collect=dict(n_sample=16),
            eval=dict(evaluator=dict(eval_freq=int(1e6))),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                       replay_buffer=dict(replay_buffer_size=1000)),
        ),
        reward_model=dict(type='rnd', obs_shape=5 * 5 * 4 + 4, hidden_size_list=[16, 16],
                          update_per_collect=1),
    ))
    create = EasyDict(dict(
        env=dict(type='minigrid_lite', import_names=['dizoo.gridworld.envs.minigrid_lite_env']),
        env_manager=dict(type='base'),
        policy=dict(type='dqn'),
    ))
    serial_pipeline_reward_model((main, create), seed=0, max_train_iter=2)


def test_memory_len_r2d2_pipeline():
    """Memory env + R2D2 (recurrent unrolls through the fused-LSTM path on
    GPU; eager here) for 2 iterations."""
    from ding.entry import serial_pipeline
    main = EasyDict(dict(
        exp_name='exp/test_memory_r2d2',
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=0.99,
                 memory_length=6),
        policy=dict(
            cuda=False, priority=True, priority_IS_weight=True,
            model=dict(obs_shape=3, action_shape=2, encoder_hidden_size_list=[16, 16], lstm_type='normal'),
            discount_factor=0.99, nstep=2, burnin_step=1, unroll_len=6, learn_unroll_len=5,
            learn=dict(update_per_collect=2, batch_size=4, learning_rate=1e-3, target_update_theta=0.01),
            collect=dict(n_sample=8, unroll_len=6, env_num=2),
            eval=dict(evaluator=dict(eval_freq=int(1e6)), env_num=2),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                       replay_buffer=dict(replay_buffer_size=1000)),
        ),
    ))
    create = EasyDict(dict(
        env=dict(type='memory_len', import_names=['dizoo.memory.envs.memory_len_env']),
        env_manager=dict(type='base'),
        policy=dict(type='r2d2'),
    ))
    serial_pipeline((main, create), seed=0, max_train_iter=2)
