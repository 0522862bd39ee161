"""BipedalWalker SAC (reference
dizoo/box2d/bipedalwalker/config/bipedalwalker_sac_config.py, stop_value 300)."""
from ding.utils import EasyDict

bipedalwalker_sac_config = EasyDict(dict(
    exp_name='bipedalwalker_sac_seed0',
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
            action_space='reparameterization',
            twin_critic=True,
        ),
        learn=dict(
            update_per_collect=64,
            batch_size=256,
            learning_rate_q=1e-3,
            learning_rate_policy=1e-3,
            learning_rate_alpha=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            auto_alpha=True,
        ),
        collect=dict(n_sample=64, unroll_len=1),
        other=dict(replay_buffer=dict(replay_buffer_size=300000, )),
    ),
))
main_config = bipedalwalker_sac_config
bipedalwalker_sac_create_config = EasyDict(dict(
    env=dict(type='bipedalwalker', import_names=['dizoo.box2d.bipedalwalker.envs.bipedalwalker_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='sac'),
))
create_config = bipedalwalker_sac_create_config
