"""CartPole SQL (reference cartpole_sql_config.py)."""
from ding.utils import EasyDict

cartpole_sql_config = EasyDict(dict(
    exp_name='cartpole_sql_seed0',
    env=dict(collector_env_num=8, evaluator_env_num=5, n_evaluator_episode=5, stop_value=195),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[128, 128, 64]),
        nstep=1,
        discount_factor=0.97,
        learn=dict(update_per_collect=3, batch_size=64, learning_rate=1e-3, alpha=0.12,
                   target_update_freq=100),
        collect=dict(n_sample=80, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=40, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
            replay_buffer=dict(replay_buffer_size=20000, ),
        ),
    ),
))
main_config = cartpole_sql_config
cartpole_sql_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='sql'),
))
create_config = cartpole_sql_create_config
