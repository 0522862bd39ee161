"""Sokoban DQN (reference dizoo/sokoban/config/sokoban_dqn_config.py,
solvable reverse-play rooms)."""
from ding.utils import EasyDict

sokoban_dqn_config = EasyDict(dict(
    exp_name='sokoban_dqn_seed0',
    env=dict(
        room_size=7,
        num_boxes=2,
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=10,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=[4, 7, 7], action_shape=4, encoder_hidden_size_list=[64, 64, 128]),
        nstep=3,
        discount_factor=0.99,
        learn=dict(update_per_collect=10, batch_size=64, learning_rate=3e-4, target_update_freq=500),
        collect=dict(n_sample=96),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(
            eps=dict(type='exp', start=1.0, end=0.05, decay=100000),
            replay_buffer=dict(replay_buffer_size=100000),
        ),
    ),
))
main_config = sokoban_dqn_config
sokoban_dqn_create_config = EasyDict(dict(
    env=dict(type='sokoban', import_names=['dizoo.sokoban.envs.sokoban_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dqn'),
))
create_config = sokoban_dqn_create_config
