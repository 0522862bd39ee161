"""MiniGrid FourRooms RND + on-policy PPO (reference
minigrid_rnd_onppo_config.py; run with serial_pipeline_reward_model)."""
from ding.utils import EasyDict

obs_dim = 13 * 13 * 4 + 4
minigrid_rnd_onppo_config = EasyDict(dict(
    exp_name='minigrid_fourrooms_rnd_onppo_seed0',
    env=dict(
        env_id='MiniGrid-FourRooms-v0',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=0.96,
    ),
    reward_model=dict(
        type='rnd',
        intrinsic_reward_type='add',
        obs_shape=obs_dim,
        hidden_size_list=[256, 256],
        learning_rate=5e-4,
        update_per_collect=10,
        batch_size=320,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        recompute_adv=True,
        model=dict(
            obs_shape=obs_dim,
            action_shape=3,
            action_space='discrete',
            encoder_hidden_size_list=[256, 128, 64],
        ),
        learn=dict(
            epoch_per_collect=10,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=200, )),
        other=dict(replay_buffer=dict(replay_buffer_size=10000, )),
    ),
))
main_config = minigrid_rnd_onppo_config
minigrid_rnd_onppo_create_config = EasyDict(dict(
    env=dict(type='minigrid', import_names=['dizoo.minigrid.envs.minigrid_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
))
create_config = minigrid_rnd_onppo_create_config
