"""Tuned PPOF preset for Pendulum-v1 (reference
ding/config/example/PPOF/gym_pendulum_v1.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='Pendulum-v1-PPOF',
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
        action_space='continuous',
        recompute_adv=True,
        model=dict(obs_shape=3, action_shape=1, action_space='continuous'),
        learn=dict(epoch_per_collect=10, batch_size=320, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.01, clip_ratio=0.2, adv_norm=True, value_norm=True),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
    ),
))
