"""POMDP (RAM Pong, corrupted obs) off-policy PPO — reference
dizoo/pomdp/config/pomdp_ppo_config.py."""
from ding.utils import EasyDict

pomdp_ppo_config = EasyDict(dict(
    exp_name='pomdp_ppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=20,
        env_id='Pong-ramNoFrameskip-v4',
        frame_stack=4,
        warp_frame=False,
        use_ram=True,
        pomdp=dict(noise_scale=0.01, zero_p=0.2, reward_noise=0.01, duplicate_p=0.2),
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        model=dict(
            obs_shape=[512, ],
            action_shape=6,
            action_space='discrete',
            encoder_hidden_size_list=[128, 128, 64],
        ),
        learn=dict(
            update_per_collect=24,
            batch_size=128,
            learning_rate=0.0001,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.1,
        ),
        collect=dict(n_sample=1024, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        other=dict(replay_buffer=dict(replay_buffer_size=10000, )),
    ),
))
main_config = pomdp_ppo_config
pomdp_ppo_create_config = EasyDict(dict(
    env=dict(type='pomdp', import_names=['dizoo.pomdp.envs.pomdp_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo_offpolicy'),
))
create_config = pomdp_ppo_create_config
