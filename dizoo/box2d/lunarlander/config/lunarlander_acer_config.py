"""LunarLander ACER (reference lunarlander_acer_config.py)."""
from ding.utils import EasyDict

lunarlander_acer_config = EasyDict(dict(
    exp_name='lunarlander_acer_seed0',
    env=dict(
        env_id='LunarLander-v2',
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=200,
    ),
    policy=dict(
        cuda=False,
        unroll_len=32,
        model=dict(obs_shape=8, action_shape=4),
        learn=dict(update_per_collect=4, batch_size=16, learning_rate=3e-4,
                   c_clip_ratio=10, trust_region=True),
        collect=dict(n_sample=64),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=5000)),
    ),
))
main_config = lunarlander_acer_config
lunarlander_acer_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='base'),
    policy=dict(type='acer'),
))
create_config = lunarlander_acer_create_config
