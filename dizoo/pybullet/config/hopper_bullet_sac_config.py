"""PyBullet HopperBulletEnv SAC (reference
dizoo/pybullet/config/hopper_bullet_sac_config.py; pybullet binaries are
unavailable offline so the mujoco-lite Hopper dynamics stand in at the
PyBullet observation shape: obs 15, act 3)."""
from ding.utils import EasyDict

hopper_bullet_sac_config = EasyDict(dict(
    exp_name='hopper_bullet_sac_seed0',
    env=dict(
        env_id='HopperBulletEnv-v0',
        obs_dim=15,
        act_dim=3,
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=2500,
    ),
    policy=dict(
        cuda=True,
        random_collect_size=10000,
        model=dict(obs_shape=15, action_shape=3, action_space='reparameterization', twin_critic=True),
        learn=dict(update_per_collect=1, batch_size=256, learning_rate_q=1e-3,
                   learning_rate_policy=1e-3, learning_rate_alpha=3e-4, target_theta=0.005,
                   discount_factor=0.99, auto_alpha=True),
        collect=dict(n_sample=1, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000)),
    ),
))
main_config = hopper_bullet_sac_config
hopper_bullet_sac_create_config = EasyDict(dict(
    env=dict(type='mujoco_lite', import_names=['dizoo.mujoco.envs.mujoco_lite_env']),
    env_manager=dict(type='base'),
    policy=dict(type='sac'),
))
create_config = hopper_bullet_sac_create_config
