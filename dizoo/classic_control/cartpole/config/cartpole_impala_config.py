"""CartPole IMPALA (reference cartpole_impala_config.py)."""
from ding.utils import EasyDict

cartpole_impala_config = EasyDict(dict(
    exp_name='cartpole_impala_seed0',
    env=dict(collector_env_num=8, evaluator_env_num=5, n_evaluator_episode=5, stop_value=195),
    policy=dict(
        cuda=False,
        unroll_len=8,
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[64, 64, 128],
                   critic_head_hidden_size=128, actor_head_hidden_size=128),
        learn=dict(update_per_collect=2, batch_size=32, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.01, discount_factor=0.9, lambda_=0.95,
                   rho_clip_ratio=1.0, c_clip_ratio=1.0),
        collect=dict(n_sample=16, ),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=1000, sliced=True)),
    ),
))
main_config = cartpole_impala_config
cartpole_impala_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='impala'),
))
create_config = cartpole_impala_create_config
