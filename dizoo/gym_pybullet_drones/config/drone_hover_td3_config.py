"""Quadrotor hover TD3 (reference
dizoo/gym_pybullet_drones/config/takeoffaviary_td3_config.py analog)."""
from ding.utils import EasyDict

drone_hover_td3_config = EasyDict(dict(
    exp_name='drone_hover_td3_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=18,
    ),
    policy=dict(
        cuda=True,
        random_collect_size=10000,
        model=dict(obs_shape=12, action_shape=4, twin_critic=True, action_space='regression'),
        learn=dict(update_per_collect=1, batch_size=256, learning_rate_actor=3e-4,
                   learning_rate_critic=3e-4, target_theta=0.005, discount_factor=0.99,
                   actor_update_freq=2, noise=True, noise_sigma=0.2,
                   noise_range=dict(min=-0.5, max=0.5)),
        collect=dict(n_sample=1, unroll_len=1, noise_sigma=0.1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000)),
    ),
))
main_config = drone_hover_td3_config
drone_hover_td3_create_config = EasyDict(dict(
    env=dict(type='drone_hover', import_names=['dizoo.gym_pybullet_drones.envs.drone_env']),
    env_manager=dict(type='base'),
    policy=dict(type='td3'),
))
create_config = drone_hover_td3_create_config
