"""Tuned TD3 preset for BipedalWalker-v3 (reference
ding/config/example/TD3/gym_bipedalwalker_v3.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='BipedalWalker-v3-TD3',
    seed=0,
    env=dict(
        type='bipedalwalker',
        import_names=['dizoo.box2d.bipedalwalker.envs.bipedalwalker_env'],
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=300,
        act_scale=True,
    ),
    policy=dict(
        cuda=True,
        random_collect_size=25000,
        model=dict(obs_shape=24, action_shape=4, action_space='regression', twin_critic=True),
        learn=dict(update_per_collect=1, batch_size=256, learning_rate_actor=1e-3, learning_rate_critic=1e-3,
                   target_theta=0.005, discount_factor=0.99, actor_update_freq=2, noise=True, noise_sigma=0.2,
                   noise_range=dict(min=-0.5, max=0.5)),
        collect=dict(n_sample=1, unroll_len=1, noise_sigma=0.1),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000)),
    ),
))
