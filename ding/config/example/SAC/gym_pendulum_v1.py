"""Tuned SAC preset for Pendulum-v1 (reference
ding/config/example/SAC/gym_pendulum_v1.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='Pendulum-v1-SAC',
    seed=0,
    env=dict(
        type='pendulum',
        import_names=['dizoo.classic_control.pendulum.envs.pendulum_env'],
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=-250,
        act_scale=True,
    ),
    policy=dict(
        cuda=True,
        random_collect_size=10000,
        model=dict(obs_shape=3, action_shape=1, action_space='reparameterization', twin_critic=True),
        learn=dict(update_per_collect=1, batch_size=256, learning_rate_q=1e-3, learning_rate_policy=1e-3,
                   learning_rate_alpha=3e-4, target_theta=0.005, discount_factor=0.99, auto_alpha=True),
        collect=dict(n_sample=1, unroll_len=1),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000)),
    ),
))
