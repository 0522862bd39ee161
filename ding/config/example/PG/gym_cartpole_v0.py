"""Tuned PG preset for CartPole-v0 (reference
ding/config/example/PG/gym_cartpole_v0.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='CartPole-v0-PG',
    seed=0,
    env=dict(
        type='cartpole',
        import_names=['dizoo.classic_control.cartpole.envs.cartpole_env'],
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=195,
        
    ),
    policy=dict(
        cuda=True,
        action_space='discrete',
        model=dict(obs_shape=4, action_shape=2, action_space='discrete'),
        learn=dict(batch_size=64, learning_rate=1e-3, entropy_weight=0.001),
        collect=dict(unroll_len=1, discount_factor=0.99),
    ),
))
