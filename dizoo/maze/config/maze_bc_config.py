"""Maze behavioral cloning (reference dizoo/maze/config/maze_bc_config.py:
obs [8, 16, 16], 4 actions, stop_value 1; expert actions come from the BFS
value-iteration sequence in ding.utils.get_vi_sequence)."""
from ding.utils import EasyDict

maze_size = 16
maze_bc_config = EasyDict(dict(
    exp_name='maze_bc_seed0',
    env=dict(
        collector_env_num=1,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        env_id='Maze',
        size=maze_size,
        stop_value=1,
    ),
    policy=dict(
        cuda=True,
        continuous=False,
        model=dict(
            obs_shape=[8, maze_size, maze_size],
            action_shape=4,
            encoder_hidden_size_list=[128, 256, 512, 1024],
        ),
        learn=dict(batch_size=32, learning_rate=0.0005, update_per_collect=1),
        collect=dict(data_type='naive', unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = maze_bc_config
maze_bc_create_config = EasyDict(dict(
    env=dict(type='maze', import_names=['dizoo.maze.envs.maze_env']),
    env_manager=dict(type='base'),
    policy=dict(type='bc'),
))
create_config = maze_bc_create_config
