"""Maze procedure cloning (BFS) — reference
dizoo/maze/config/maze_pc_config.py: supervised training on value-iteration
BFS traces from ding.utils.get_vi_sequence; run with
ding.entry.serial_pipeline_pc."""
from ding.utils import EasyDict

maze_size = 16
maze_pc_config = EasyDict(dict(
    exp_name='maze_pc_seed0',
    env=dict(
        collector_env_num=1,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        size=maze_size,
        stop_value=1,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=[8, maze_size, maze_size], action_shape=4,
                   encoder_hidden_size_list=[128, 256, 512]),
        learn=dict(batch_size=32, learning_rate=5e-4, train_epoch=100),
        collect=dict(unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = maze_pc_config
maze_pc_create_config = EasyDict(dict(
    env=dict(type='maze', import_names=['dizoo.maze.envs.maze_env']),
    env_manager=dict(type='base'),
    policy=dict(type='pc_bfs'),
))
create_config = maze_pc_create_config
