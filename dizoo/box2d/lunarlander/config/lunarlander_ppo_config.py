"""LunarLander on-policy PPO (reference
dizoo/box2d/lunarlander/config/lunarlander_ppo_config.py, stop_value 200)."""
from ding.utils import EasyDict

lunarlander_ppo_config = EasyDict(dict(
    exp_name='lunarlander_ppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        env_id='LunarLander-v2',
        n_evaluator_episode=8,
        stop_value=200,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        recompute_adv=True,
        model=dict(
            obs_shape=8,
            action_shape=4,
            action_space='discrete',
        ),
        learn=dict(
            epoch_per_collect=8,
            batch_size=800,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=1600, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=100, )),
    ),
))
main_config = lunarlander_ppo_config
lunarlander_ppo_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
))
create_config = lunarlander_ppo_create_config
