"""MiniGrid Empty-8x8 on-policy PPO (reference
dizoo/minigrid/config/minigrid_onppo_config.py)."""
from ding.utils import EasyDict

obs_dim = 8 * 8 * 4 + 4
minigrid_onppo_config = EasyDict(dict(
    exp_name='minigrid_empty8_onppo_seed0',
    env=dict(
        env_id='MiniGrid-Empty-8x8-v0',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=0.96,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        recompute_adv=True,
        model=dict(
            obs_shape=obs_dim,
            action_shape=3,
            action_space='discrete',
            encoder_hidden_size_list=[256, 128, 64],
        ),
        learn=dict(
            epoch_per_collect=10,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=200, )),
    ),
))
main_config = minigrid_onppo_config
minigrid_onppo_create_config = EasyDict(dict(
    env=dict(type='minigrid', import_names=['dizoo.minigrid.envs.minigrid_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
))
create_config = minigrid_onppo_create_config
