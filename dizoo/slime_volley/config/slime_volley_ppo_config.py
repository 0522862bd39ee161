"""Slime Volleyball vs-bot on-policy PPO (reference
dizoo/slime_volley/config/slime_volley_ppo_config.py, stop_value 5)."""
from ding.utils import EasyDict

slime_volley_ppo_config = EasyDict(dict(
    exp_name='slime_volley_ppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        agent_vs_agent=False,
        stop_value=5,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        recompute_adv=True,
        model=dict(
            obs_shape=12,
            action_shape=6,
            action_space='discrete',
            encoder_hidden_size_list=[64, 64],
        ),
        learn=dict(
            epoch_per_collect=5,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=500, )),
    ),
))
main_config = slime_volley_ppo_config
slime_volley_ppo_create_config = EasyDict(dict(
    env=dict(type='slime_volley', import_names=['dizoo.slime_volley.envs.slime_volley_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
))
create_config = slime_volley_ppo_create_config
