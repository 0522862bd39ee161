"""Pendulum TD3+BC offline (reference pendulum_td3_bc_config.py). Generate
the dataset with the offline_gen pipeline (collect with pendulum_td3)."""
from ding.utils import EasyDict

pendulum_td3_bc_config = EasyDict(dict(
    exp_name='pendulum_td3_bc_seed0',
    env=dict(collector_env_num=8, evaluator_env_num=5, n_evaluator_episode=5,
             stop_value=-250, act_scale=True),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=3, action_shape=1, action_space='regression', twin_critic=True),
        learn=dict(train_epoch=3, batch_size=128, learning_rate_actor=3e-4,
                   learning_rate_critic=3e-4, alpha=2.5, target_theta=0.005,
                   discount_factor=0.99, actor_update_freq=2, noise=True,
                   noise_sigma=0.2, noise_range=dict(min=-0.5, max=0.5),
                   normalize_states=True),
        collect=dict(data_type='naive', data_path=None, unroll_len=1, noise_sigma=0.1),
        eval=dict(evaluator=dict(eval_freq=100, )),
    ),
))
main_config = pendulum_td3_bc_config
pendulum_td3_bc_create_config = EasyDict(dict(
    env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
    env_manager=dict(type='base'),
    policy=dict(type='td3_bc'),
))
create_config = pendulum_td3_bc_create_config
