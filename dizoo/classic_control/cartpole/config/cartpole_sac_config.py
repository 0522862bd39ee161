"""CartPole discrete SAC (reference cartpole_sac_config.py)."""
from ding.utils import EasyDict

cartpole_sac_config = EasyDict(dict(
    exp_name='cartpole_sac_seed0',
    env=dict(collector_env_num=8, evaluator_env_num=5, n_evaluator_episode=5, stop_value=195),
    policy=dict(
        cuda=False,
        random_collect_size=0,
        multi_agent=False,
        model=dict(obs_shape=4, action_shape=2, twin_critic=True,
                   actor_head_hidden_size=64, critic_head_hidden_size=64),
        learn=dict(update_per_collect=2, batch_size=64, learning_rate_q=1e-3,
                   learning_rate_policy=1e-3, learning_rate_alpha=3e-4, target_theta=0.005,
                   discount_factor=0.99, auto_alpha=False),
        collect=dict(n_sample=80, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=20000, )),
    ),
))
main_config = cartpole_sac_config
cartpole_sac_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='discrete_sac'),
))
create_config = cartpole_sac_create_config
