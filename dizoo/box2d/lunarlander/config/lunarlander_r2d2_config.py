"""LunarLander R2D2 (reference lunarlander_r2d2_config.py)."""
from ding.utils import EasyDict

lunarlander_r2d2_config = EasyDict(dict(
    exp_name='lunarlander_r2d2_seed0',
    env=dict(
        env_id='LunarLander-v2',
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=200,
    ),
    policy=dict(
        cuda=False,
        priority=True,
        priority_IS_weight=True,
        model=dict(obs_shape=8, action_shape=4, encoder_hidden_size_list=[128, 128, 64],
                   lstm_type='normal'),
        discount_factor=0.997,
        nstep=5,
        burnin_step=2,
        unroll_len=40,
        learn_unroll_len=38,
        learn=dict(update_per_collect=8, batch_size=64, learning_rate=5e-4, target_update_theta=0.001),
        collect=dict(n_sample=32, unroll_len=40, env_num=8),
        eval=dict(evaluator=dict(eval_freq=100, ), env_num=5),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.05, decay=10000),
            replay_buffer=dict(replay_buffer_size=10000, ),
        ),
    ),
))
main_config = lunarlander_r2d2_config
lunarlander_r2d2_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='base'),
    policy=dict(type='r2d2'),
))
create_config = lunarlander_r2d2_create_config
