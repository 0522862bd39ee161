"""CartPole A2C (reference cartpole_a2c_config.py)."""
from ding.utils import EasyDict

cartpole_a2c_config = EasyDict(dict(
    exp_name='cartpole_a2c_seed0',
    env=dict(collector_env_num=8, evaluator_env_num=5, n_evaluator_episode=5, stop_value=195),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[64, 64, 128]),
        learn=dict(batch_size=64, learning_rate=1e-3, value_weight=0.5, entropy_weight=0.01),
        collect=dict(n_sample=64, discount_factor=0.9, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=100, )),
    ),
))
main_config = cartpole_a2c_config
cartpole_a2c_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='a2c'),
))
create_config = cartpole_a2c_create_config
