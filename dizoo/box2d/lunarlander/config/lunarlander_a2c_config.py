"""LunarLander A2C (reference lunarlander_a2c_config.py, stop_value 200)."""
from ding.utils import EasyDict

lunarlander_a2c_config = EasyDict(dict(
    exp_name='lunarlander_a2c_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        env_id='LunarLander-v2',
        n_evaluator_episode=8,
        stop_value=200,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        model=dict(obs_shape=8, action_shape=4, action_space='discrete'),
        learn=dict(
            batch_size=160,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            adv_norm=True,
        ),
        collect=dict(n_sample=320, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
    ),
))
main_config = lunarlander_a2c_config
lunarlander_a2c_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='a2c'),
))
create_config = lunarlander_a2c_create_config
