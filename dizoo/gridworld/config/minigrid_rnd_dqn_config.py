"""MiniGrid-lite DQN + RND exploration config (pipeline:
serial_pipeline_reward_model)."""
from ding.utils import EasyDict

obs_dim = 8 * 8 * 4 + 4

main_config = EasyDict(dict(
    exp_name='minigrid_rnd_dqn',
    env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=0.9,
             grid_size=8),
    policy=dict(
        cuda=True, nstep=3, discount_factor=0.99,
        model=dict(obs_shape=obs_dim, action_shape=3, encoder_hidden_size_list=[128, 128, 64]),
        learn=dict(update_per_collect=5, batch_size=64, learning_rate=1e-3, target_update_freq=100),
        collect=dict(n_sample=64),
        eval=dict(evaluator=dict(eval_freq=200)),
        other=dict(eps=dict(type='exp', start=0.95, end=0.05, decay=50000),
                   replay_buffer=dict(replay_buffer_size=100000)),
    ),
    reward_model=dict(type='rnd', obs_shape=obs_dim, hidden_size_list=[64, 64], update_per_collect=4),
))

create_config = EasyDict(dict(
    env=dict(type='minigrid_lite', import_names=['dizoo.gridworld.envs.minigrid_lite_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dqn'),
))
