"""Pendulum continuous PPO (reference pendulum_ppo_config.py)."""
from ding.utils import EasyDict

pendulum_ppo_config = EasyDict(dict(
    exp_name='pendulum_ppo_seed0',
    env=dict(collector_env_num=8, evaluator_env_num=5, n_evaluator_episode=5,
             stop_value=-250, act_scale=True),
    policy=dict(
        cuda=False,
        recompute_adv=True,
        action_space='continuous',
        model=dict(obs_shape=3, action_shape=1, action_space='continuous',
                   encoder_hidden_size_list=[64, 64],
                   actor_head_hidden_size=64, critic_head_hidden_size=64),
        learn=dict(epoch_per_collect=10, batch_size=32, learning_rate=3e-4,
                   value_weight=0.5, entropy_weight=0.0, clip_ratio=0.2,
                   adv_norm=True, value_norm=True),
        collect=dict(n_sample=200, unroll_len=1, discount_factor=0.9, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=200, )),
    ),
))
main_config = pendulum_ppo_config
pendulum_ppo_create_config = EasyDict(dict(
    env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = pendulum_ppo_create_config
