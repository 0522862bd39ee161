"""LunarLander SQL (reference lunarlander_sql_config.py)."""
from ding.utils import EasyDict

lunarlander_sql_config = EasyDict(dict(
    exp_name='lunarlander_sql_seed0',
    env=dict(
        env_id='LunarLander-v2',
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=200,
    ),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=8, action_shape=4, encoder_hidden_size_list=[128, 128, 64]),
        nstep=1,
        discount_factor=0.97,
        learn=dict(update_per_collect=3, batch_size=64, learning_rate=1e-3, alpha=0.12,
                   target_update_freq=100),
        collect=dict(n_sample=80, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=40, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
            replay_buffer=dict(replay_buffer_size=20000, ),
        ),
    ),
))
main_config = lunarlander_sql_config
lunarlander_sql_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='base'),
    policy=dict(type='sql'),
))
create_config = lunarlander_sql_create_config
