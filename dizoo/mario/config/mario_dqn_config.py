"""Mario-lite DQN (reference dizoo/mario/mario_dqn_config.py)."""
from ding.utils import EasyDict

mario_dqn_config = EasyDict(dict(
    exp_name='mario_dqn_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=200,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=[4, 84, 84], action_shape=3, encoder_hidden_size_list=[128, 128, 512]),
        nstep=3,
        discount_factor=0.99,
        learn=dict(update_per_collect=10, batch_size=32, learning_rate=1e-4, target_update_freq=500),
        collect=dict(n_sample=96),
        eval=dict(evaluator=dict(eval_freq=2000, )),
        other=dict(
            eps=dict(type='exp', start=1., end=0.05, decay=250000),
            replay_buffer=dict(replay_buffer_size=100000),
        ),
    ),
))
main_config = mario_dqn_config
mario_dqn_create_config = EasyDict(dict(
    env=dict(type='mario', import_names=['dizoo.mario.envs.mario_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='dqn'),
))
create_config = mario_dqn_create_config
