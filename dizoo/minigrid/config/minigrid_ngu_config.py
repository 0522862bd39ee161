"""MiniGrid FourRooms NGU (reference minigrid_ngu_config.py; R2D2 learner +
RND lifelong x episodic novelty, run with serial_pipeline_ngu)."""
from ding.utils import EasyDict

obs_dim = 13 * 13 * 4 + 4
minigrid_ngu_config = EasyDict(dict(
    exp_name='minigrid_fourrooms_ngu_seed0',
    env=dict(
        env_id='MiniGrid-FourRooms-v0',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=0.96,
    ),
    rnd_reward_model=dict(
        type='rnd-ngu',
        intrinsic_reward_type='add',
        obs_shape=obs_dim,
        hidden_size_list=[128, 64],
        learning_rate=5e-4,
        update_per_collect=10,
        batch_size=320,
    ),
    episodic_reward_model=dict(
        type='episodic',
        intrinsic_reward_type='add',
        obs_shape=obs_dim,
        action_shape=3,
        hidden_size_list=[128, 64],
        learning_rate=5e-4,
        update_per_collect=10,
        batch_size=320,
    ),
    policy=dict(
        cuda=False,
        priority=True,
        priority_IS_weight=True,
        model=dict(
            obs_shape=obs_dim,
            action_shape=3,
            encoder_hidden_size_list=[128, 128, 64],
            lstm_type='normal',
        ),
        discount_factor=0.997,
        nstep=5,
        burnin_step=2,
        unroll_len=40,
        learn_unroll_len=38,
        learn=dict(
            update_per_collect=8,
            batch_size=64,
            learning_rate=5e-4,
            target_update_theta=0.001,
        ),
        collect=dict(n_sample=32, unroll_len=40, env_num=8),
        eval=dict(evaluator=dict(eval_freq=200, ), env_num=5),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.05, decay=100000),
            replay_buffer=dict(replay_buffer_size=10000, ),
        ),
    ),
))
main_config = minigrid_ngu_config
minigrid_ngu_create_config = EasyDict(dict(
    env=dict(type='minigrid', import_names=['dizoo.minigrid.envs.minigrid_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ngu'),
))
create_config = minigrid_ngu_create_config
