"""Tuned DDPG preset for Pendulum-v1 (reference
ding/config/example/DDPG/gym_pendulum_v1.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='Pendulum-v1-DDPG',
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
        random_collect_size=25000,
        model=dict(obs_shape=3, action_shape=1, action_space='regression', twin_critic=False),
        learn=dict(update_per_collect=1, batch_size=256, learning_rate_actor=1e-3, learning_rate_critic=1e-3,
                   target_theta=0.005, discount_factor=0.99, actor_update_freq=1, noise=False),
        collect=dict(n_sample=1, unroll_len=1, noise_sigma=0.1),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000)),
    ),
))
