"""Pendulum TD3-VAE (HyAR-style latent action space; reference
dizoo/classic_control/pendulum/config/pendulum_td3_vae_config.py; run with
ding.entry.serial_pipeline_td3_vae)."""
from ding.utils import EasyDict

pendulum_td3_vae_config = EasyDict(dict(
    exp_name='pendulum_td3_vae_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=-250,
        act_scale=True,
    ),
    policy=dict(
        cuda=False,
        random_collect_size=1000,
        original_action_shape=1,
        model=dict(obs_shape=3, action_shape=2, action_space='regression', twin_critic=True),
        learn=dict(
            warm_up_update=64,
            rl_vae_update_circle=1,
            update_per_collect_rl=2,
            update_per_collect_vae=1,
            batch_size=128,
            learning_rate_actor=3e-4,
            learning_rate_critic=3e-4,
            learning_rate_vae=1e-4,
            target_theta=0.005,
            discount_factor=0.99,
            actor_update_freq=2,
            noise=True,
            noise_sigma=0.1,
            noise_range=dict(min=-0.5, max=0.5),
        ),
        collect=dict(n_sample=48, unroll_len=1, noise_sigma=0.1),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=20000)),
    ),
))
main_config = pendulum_td3_vae_config
pendulum_td3_vae_create_config = EasyDict(dict(
    env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
    env_manager=dict(type='base'),
    policy=dict(type='td3_vae'),
))
create_config = pendulum_td3_vae_create_config
