"""Tuned SAC preset for Hopper-v3 (reference
ding/config/example/SAC/gym_hopper_v3.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='Hopper-v3-SAC',
    seed=0,
    env=dict(
        type='mujoco_lite',
        import_names=['dizoo.mujoco.envs.mujoco_lite_env'],
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=6000,
        env_id='Hopper-v3',
    ),
    policy=dict(
        cuda=True,
        random_collect_size=10000,
        model=dict(obs_shape=11, action_shape=3, action_space='reparameterization', twin_critic=True),
        learn=dict(update_per_collect=1, batch_size=256, learning_rate_q=1e-3, learning_rate_policy=1e-3,
                   learning_rate_alpha=3e-4, target_theta=0.005, discount_factor=0.99, auto_alpha=True),
        collect=dict(n_sample=1, unroll_len=1),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000)),
    ),
))
