"""Procgen maze DQN (reference dizoo/procgen/config/maze_dqn_config.py,
stop_value 10, obs [3,64,64], 15 actions)."""
from ding.utils import EasyDict

maze_dqn_config = EasyDict(dict(
    exp_name='maze_dqn_seed0',
    env=dict(
        env_id='maze',
        collector_env_num=4,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=10,
    ),
    policy=dict(
        cuda=True,
        model=dict(
            obs_shape=[3, 64, 64],
            action_shape=15,
            encoder_hidden_size_list=[128, 128, 512],
        ),
        discount_factor=0.99,
        nstep=1,
        learn=dict(
            update_per_collect=10,
            batch_size=32,
            learning_rate=0.0005,
            target_update_freq=500,
        ),
        collect=dict(n_sample=100),
        eval=dict(evaluator=dict(eval_freq=5000, )),
        other=dict(
            eps=dict(type='exp', start=1., end=0.05, decay=250000),
            replay_buffer=dict(replay_buffer_size=100000, ),
        ),
    ),
))
main_config = maze_dqn_config
maze_dqn_create_config = EasyDict(dict(
    env=dict(type='procgen', import_names=['dizoo.procgen.envs.procgen_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='dqn'),
))
create_config = maze_dqn_create_config
