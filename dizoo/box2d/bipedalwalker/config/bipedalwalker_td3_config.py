"""BipedalWalker TD3 (reference bipedalwalker_td3_config.py, stop_value 300)."""
from ding.utils import EasyDict

bipedalwalker_td3_config = EasyDict(dict(
    exp_name='bipedalwalker_td3_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=300,
        act_scale=True,
    ),
    policy=dict(
        cuda=False,
        random_collect_size=10000,
        model=dict(
            obs_shape=24,
            action_shape=4,
            twin_critic=True,
            action_space='regression',
        ),
        learn=dict(
            update_per_collect=64,
            batch_size=256,
            learning_rate_actor=3e-4,
            learning_rate_critic=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            actor_update_freq=2,
            noise=True,
            noise_sigma=0.2,
            noise_range=dict(min=-0.5, max=0.5),
        ),
        collect=dict(n_sample=64, unroll_len=1, noise_sigma=0.1),
        other=dict(replay_buffer=dict(replay_buffer_size=300000, )),
    ),
))
main_config = bipedalwalker_td3_config
bipedalwalker_td3_create_config = EasyDict(dict(
    env=dict(type='bipedalwalker', import_names=['dizoo.box2d.bipedalwalker.envs.bipedalwalker_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='td3'),
))
create_config = bipedalwalker_td3_create_config
