"""Pendulum SQIL-SAC (sparse imitation: expert transitions labelled r=1,
agent r=0; reference pendulum_sqil_sac_config.py; run with
serial_pipeline_sqil + an expert SAC config)."""
from ding.utils import EasyDict

pendulum_sqil_sac_config = EasyDict(dict(
    exp_name='pendulum_sqil_sac_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=-250,
        act_scale=True,
    ),
    policy=dict(
        cuda=False,
        random_collect_size=1000,
        expert_random_collect_size=1000,
        model=dict(obs_shape=3, action_shape=1, action_space='reparameterization', twin_critic=True),
        learn=dict(
            update_per_collect=2,
            batch_size=128,
            learning_rate_q=1e-3,
            learning_rate_policy=1e-3,
            learning_rate_alpha=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            auto_alpha=True,
        ),
        collect=dict(n_sample=10, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=100000)),
    ),
))
main_config = pendulum_sqil_sac_config
pendulum_sqil_sac_create_config = EasyDict(dict(
    env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
    env_manager=dict(type='base'),
    policy=dict(type='sqil_sac'),
))
create_config = pendulum_sqil_sac_create_config
