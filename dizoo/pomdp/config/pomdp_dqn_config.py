"""POMDP (RAM Pong, corrupted obs) DQN — reference
dizoo/pomdp/config/pomdp_dqn_config.py (obs [512], noise/zero/duplicate)."""
from ding.utils import EasyDict

pomdp_dqn_config = EasyDict(dict(
    exp_name='pomdp_dqn_seed0',
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
        priority=False,
        model=dict(
            obs_shape=[512, ],
            action_shape=6,
            encoder_hidden_size_list=[128, 128, 512],
        ),
        nstep=3,
        discount_factor=0.99,
        learn=dict(
            update_per_collect=10,
            batch_size=32,
            learning_rate=0.0001,
            target_update_freq=500,
        ),
        collect=dict(n_sample=100, ),
        eval=dict(evaluator=dict(eval_freq=4000, )),
        other=dict(
            eps=dict(type='exp', start=1., end=0.05, decay=250000),
            replay_buffer=dict(replay_buffer_size=100000, ),
        ),
    ),
))
main_config = pomdp_dqn_config
pomdp_dqn_create_config = EasyDict(dict(
    env=dict(type='pomdp', import_names=['dizoo.pomdp.envs.pomdp_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='dqn'),
))
create_config = pomdp_dqn_create_config
