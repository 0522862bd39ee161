"""CartPole on-policy PPO + ST-DIM auxiliary loss (reference cartpole_ppo_stdim_config.py)."""
from ding.utils import EasyDict

cartpole_ppo_stdim_config = EasyDict(dict(
    exp_name='cartpole_ppo_stdim_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=195,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        recompute_adv=True,
        aux_loss_weight=0.003,
        model=dict(obs_shape=4, action_shape=2, action_space='discrete'),
        learn=dict(epoch_per_collect=2, batch_size=64, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.01, clip_ratio=0.2, adv_norm=True, value_norm=True),
        collect=dict(n_sample=256, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=100, )),
    ),
))
main_config = cartpole_ppo_stdim_config
cartpole_ppo_stdim_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo_stdim'),
))
create_config = cartpole_ppo_stdim_create_config
