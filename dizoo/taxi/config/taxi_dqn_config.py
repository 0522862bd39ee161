"""Taxi DQN (reference dizoo/taxi/config/taxi_dqn_config.py: obs 34 encoded,
6 actions, stop_value 20)."""
from ding.utils import EasyDict

taxi_dqn_config = EasyDict(dict(
    exp_name='taxi_dqn_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        env_id='Taxi-v3',
        max_episode_steps=60,
        stop_value=20,
    ),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=34, action_shape=6, encoder_hidden_size_list=[128, 128]),
        nstep=3,
        discount_factor=0.99,
        learn=dict(
            update_per_collect=10,
            batch_size=64,
            learning_rate=0.0001,
            target_update_freq=500,
        ),
        collect=dict(n_sample=32),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(
            eps=dict(type='exp', start=1, end=0.05, decay=3000000),
            replay_buffer=dict(replay_buffer_size=100000, ),
        ),
    ),
))
main_config = taxi_dqn_config
taxi_dqn_create_config = EasyDict(dict(
    env=dict(type='taxi', import_names=['dizoo.taxi.envs.taxi_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='dqn'),
))
create_config = taxi_dqn_create_config
