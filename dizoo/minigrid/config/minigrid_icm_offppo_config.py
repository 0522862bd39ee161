"""MiniGrid DoorKey ICM + off-policy PPO (reference
minigrid_icm_offppo_config.py; run with serial_pipeline_reward_model)."""
from ding.utils import EasyDict

obs_dim = 8 * 8 * 4 + 4
minigrid_icm_offppo_config = EasyDict(dict(
    exp_name='minigrid_doorkey_icm_offppo_seed0',
    env=dict(
        env_id='MiniGrid-DoorKey-8x8-v0',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=0.96,
    ),
    reward_model=dict(
        type='icm',
        intrinsic_reward_type='add',
        obs_shape=obs_dim,
        action_shape=3,
        hidden_size_list=[256, 64],
        learning_rate=1e-3,
        update_per_collect=10,
        batch_size=320,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        model=dict(
            obs_shape=obs_dim,
            action_shape=3,
            action_space='discrete',
            encoder_hidden_size_list=[256, 128, 64],
        ),
        learn=dict(
            update_per_collect=10,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            clip_ratio=0.2,
            adv_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=200, )),
        other=dict(replay_buffer=dict(replay_buffer_size=100000, )),
    ),
))
main_config = minigrid_icm_offppo_config
minigrid_icm_offppo_create_config = EasyDict(dict(
    env=dict(type='minigrid', import_names=['dizoo.minigrid.envs.minigrid_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo_offpolicy'),
))
create_config = minigrid_icm_offppo_create_config
