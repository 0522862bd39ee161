"""LunarLander IMPALA (reference lunarlander_impala_config.py, stop_value 200)."""
from ding.utils import EasyDict

lunarlander_impala_config = EasyDict(dict(
    exp_name='lunarlander_impala_seed0',
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
        unroll_len=32,
        model=dict(obs_shape=8, action_shape=4),
        learn=dict(
            update_per_collect=2,
            batch_size=64,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            discount_factor=0.99,
            lambda_=0.95,
        ),
        collect=dict(n_sample=64),
        other=dict(replay_buffer=dict(replay_buffer_size=1000, )),
    ),
))
main_config = lunarlander_impala_config
lunarlander_impala_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='impala'),
))
create_config = lunarlander_impala_create_config
