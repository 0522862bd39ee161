"""bsuite deep_sea/10 DQN (exploration diagnostic; reference
dizoo/bsuite/config/serial/deep_sea/deep_sea_dqn_config.py)."""
from ding.utils import EasyDict

deep_sea_dqn_config = EasyDict(dict(
    exp_name='bsuite_deep_sea_dqn_seed0',
    env=dict(
        env_id='deep_sea/10',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=10,
        stop_value=0.99,
    ),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=100, action_shape=2, encoder_hidden_size_list=[128, 128, 64], dueling=True),
        nstep=1,
        discount_factor=0.99,
        learn=dict(update_per_collect=10, batch_size=64, learning_rate=1e-3, target_update_freq=100),
        collect=dict(n_sample=64),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.05, decay=50000),
            replay_buffer=dict(replay_buffer_size=100000),
        ),
    ),
))
main_config = deep_sea_dqn_config
deep_sea_dqn_create_config = EasyDict(dict(
    env=dict(type='bsuite', import_names=['dizoo.bsuite.envs.bsuite_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dqn'),
))
create_config = deep_sea_dqn_create_config
