"""Tuned SQL preset for LunarLander-v2 (reference
ding/config/example/SQL/gym_lunarlander_v2.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='LunarLander-v2-SQL',
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
        discount_factor=0.97,
        nstep=1,
        model=dict(obs_shape=8, action_shape=4, encoder_hidden_size_list=[128, 128, 64]),
        learn=dict(update_per_collect=5, batch_size=64, learning_rate=1e-3, alpha=0.12),
        collect=dict(n_sample=8, unroll_len=1),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
            replay_buffer=dict(replay_buffer_size=20000),
        ),
    ),
))
