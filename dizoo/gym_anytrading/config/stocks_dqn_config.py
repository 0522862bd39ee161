"""gym-anytrading stocks DQN (reference
dizoo/gym_anytrading/config/stocks_dqn_config.py)."""
from ding.utils import EasyDict

stocks_dqn_config = EasyDict(dict(
    exp_name='stocks_dqn_seed0',
    env=dict(
        env_id='stocks-v0',
        window_size=20,
        eps_length=200,
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=2,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=40, action_shape=2, encoder_hidden_size_list=[128, 128, 64]),
        nstep=3,
        discount_factor=0.99,
        learn=dict(update_per_collect=10, batch_size=64, learning_rate=1e-4, target_update_freq=500),
        collect=dict(n_sample=64),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=50000),
            replay_buffer=dict(replay_buffer_size=100000),
        ),
    ),
))
main_config = stocks_dqn_config
stocks_dqn_create_config = EasyDict(dict(
    env=dict(type='stocks', import_names=['dizoo.gym_anytrading.envs.stocks_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dqn'),
))
create_config = stocks_dqn_create_config
