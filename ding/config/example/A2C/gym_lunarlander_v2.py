"""Tuned A2C preset for LunarLander-v2 (reference
ding/config/example/A2C/gym_lunarlander_v2.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='LunarLander-v2-A2C',
    seed=0,
    env=dict(
        type='lunarlander',
        import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env'],
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=200,
        env_id='LunarLander-v2',
    ),
    policy=dict(
        cuda=True,
        action_space='discrete',
        model=dict(obs_shape=8, action_shape=4, action_space='discrete'),
        learn=dict(batch_size=160, learning_rate=3e-4, value_weight=0.5, entropy_weight=0.01, adv_norm=True),
        collect=dict(n_sample=320, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
    ),
))
