"""Competitive Pong vs builtin bot, DQN (reference
dizoo/competitive_rl/config/cpong_dqn_default_config.py)."""
from ding.utils import EasyDict

cpong_dqn_config = EasyDict(dict(
    exp_name='cpong_dqn_seed0',
    env=dict(
        opponent='builtin',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=5,
    ),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=8, action_shape=3, encoder_hidden_size_list=[128, 128, 64], dueling=True),
        nstep=3,
        discount_factor=0.99,
        learn=dict(update_per_collect=10, batch_size=64, learning_rate=3e-4, target_update_freq=500),
        collect=dict(n_sample=96),
        eval=dict(evaluator=dict(eval_freq=500, )),
        other=dict(
            eps=dict(type='exp', start=1.0, end=0.05, decay=100000),
            replay_buffer=dict(replay_buffer_size=100000),
        ),
    ),
))
main_config = cpong_dqn_config
cpong_dqn_create_config = EasyDict(dict(
    env=dict(type='competitive_pong', import_names=['dizoo.competitive_rl.envs.cpong_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dqn'),
))
create_config = cpong_dqn_create_config
