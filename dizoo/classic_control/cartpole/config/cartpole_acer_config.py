"""CartPole ACER (reference cartpole_acer_config.py)."""
from ding.utils import EasyDict

cartpole_acer_config = EasyDict(dict(
    exp_name='cartpole_acer_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=195,
    ),
    policy=dict(
        cuda=False,
        unroll_len=32,
        model=dict(obs_shape=4, action_shape=2),
        learn=dict(update_per_collect=4, batch_size=16, learning_rate=3e-4,
                   c_clip_ratio=10, trust_region=True),
        collect=dict(n_sample=64),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=5000)),
    ),
))
main_config = cartpole_acer_config
cartpole_acer_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='acer'),
))
create_config = cartpole_acer_create_config
