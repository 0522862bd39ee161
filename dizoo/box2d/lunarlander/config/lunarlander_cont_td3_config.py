"""LunarLanderContinuous TD3 (reference lunarlander_cont_td3_config.py)."""
from ding.utils import EasyDict

lunarlander_cont_td3_config = EasyDict(dict(
    exp_name='lunarlander_cont_td3_seed0',
    env=dict(
        env_id='LunarLanderContinuous-v2',
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=200,
        act_scale=True,
    ),
    policy=dict(
        cuda=False,
        random_collect_size=10000,
        model=dict(
            obs_shape=8,
            action_shape=2,
            twin_critic=True,
            action_space='regression',
        ),
        learn=dict(
            update_per_collect=256,
            batch_size=128,
            learning_rate_actor=3e-4,
            learning_rate_critic=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            actor_update_freq=2,
            noise=True,
            noise_sigma=0.1,
            noise_range=dict(min=-0.5, max=0.5),
        ),
        collect=dict(n_sample=256, unroll_len=1, noise_sigma=0.1),
        other=dict(replay_buffer=dict(replay_buffer_size=100000, )),
    ),
))
main_config = lunarlander_cont_td3_config
lunarlander_cont_td3_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='td3'),
))
create_config = lunarlander_cont_td3_create_config
