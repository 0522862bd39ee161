"""CartPole PPO (reference cartpole_ppo_config.py — the canonical
on-policy hello-world)."""
from ding.utils import EasyDict

cartpole_ppo_config = EasyDict(dict(
    exp_name='cartpole_ppo_seed0',
    env=dict(collector_env_num=8, evaluator_env_num=5, n_evaluator_episode=5, stop_value=195),
    policy=dict(
        cuda=False,
        action_space='discrete',
        model=dict(obs_shape=4, action_shape=2, action_space='discrete',
                   encoder_hidden_size_list=[64, 64, 128], critic_head_hidden_size=128,
                   actor_head_hidden_size=128),
        learn=dict(epoch_per_collect=2, batch_size=64, learning_rate=1e-3, value_weight=0.5,
                   entropy_weight=0.01, clip_ratio=0.2, adv_norm=True, value_norm=True),
        collect=dict(n_sample=256, unroll_len=1, discount_factor=0.9, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=100, )),
    ),
))
main_config = cartpole_ppo_config
cartpole_ppo_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = cartpole_ppo_create_config
