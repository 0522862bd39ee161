"""CartPole off-policy PPG (reference cartpole_ppg_config.py)."""
from ding.utils import EasyDict

cartpole_ppg_offpolicy_config = EasyDict(dict(
    exp_name='cartpole_ppg_offpolicy_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=195,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        model=dict(obs_shape=4, action_shape=2, action_space='discrete',
                   encoder_hidden_size_list=[64, 64, 128]),
        learn=dict(update_per_collect=2, batch_size=64, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.01, clip_ratio=0.2, aux_freq=4, aux_train_epoch=2),
        collect=dict(n_sample=256, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(
            multi_buffer=True,
            policy=dict(replay_buffer_size=1000),
            value=dict(replay_buffer_size=4000),
        )),
    ),
))
main_config = cartpole_ppg_offpolicy_config
cartpole_ppg_offpolicy_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppg_offpolicy'),
))
create_config = cartpole_ppg_offpolicy_create_config
