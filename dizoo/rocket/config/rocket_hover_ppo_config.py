"""Rocket hover on-policy PPO (reference
dizoo/rocket/config/rocket_hover_ppo_config.py)."""
from ding.utils import EasyDict

rocket_hover_ppo_config = EasyDict(dict(
    exp_name='rocket_hover_ppo_seed0',
    env=dict(
        task='hover',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=8,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        recompute_adv=True,
        model=dict(obs_shape=8, action_shape=9, action_space='discrete'),
        learn=dict(epoch_per_collect=10, batch_size=320, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.01, clip_ratio=0.2, adv_norm=True, value_norm=True),
        collect=dict(n_sample=2048, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=300, )),
    ),
))
main_config = rocket_hover_ppo_config
rocket_hover_ppo_create_config = EasyDict(dict(
    env=dict(type='rocket', import_names=['dizoo.rocket.envs.rocket_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = rocket_hover_ppo_create_config
