"""Tuned PG preset for Pendulum-v1 (reference
ding/config/example/PG/gym_pendulum_v1.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='Pendulum-v1-PG',
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
        model=dict(obs_shape=3, action_shape=1, action_space='continuous'),
        learn=dict(batch_size=64, learning_rate=1e-3, entropy_weight=0.001),
        collect=dict(unroll_len=1, discount_factor=0.99),
    ),
))
