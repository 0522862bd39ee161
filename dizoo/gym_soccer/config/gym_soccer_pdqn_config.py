"""Half-field soccer PDQN over hybrid actions (reference
dizoo/gym_soccer/config/gym_soccer_pdqn_config.py)."""
from ding.utils import EasyDict

gym_soccer_pdqn_config = EasyDict(dict(
    exp_name='gym_soccer_pdqn_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=0.95,
    ),
    policy=dict(
        cuda=False,
        discount_factor=0.99,
        nstep=1,
        model=dict(
            obs_shape=10,
            action_shape=dict(action_type_shape=3, action_args_shape=2),
        ),
        learn=dict(update_per_collect=100, batch_size=320, learning_rate_dis=3e-4,
                   learning_rate_cont=3e-4, target_theta=0.001, update_circle=10),
        collect=dict(n_sample=3200, unroll_len=1, noise_sigma=0.1),
        other=dict(
            eps=dict(type='exp', start=1, end=0.1, decay=100000),
            replay_buffer=dict(replay_buffer_size=1000000, ),
        ),
    ),
))
main_config = gym_soccer_pdqn_config
gym_soccer_pdqn_create_config = EasyDict(dict(
    env=dict(type='gym_soccer', import_names=['dizoo.gym_soccer.envs.soccer_env']),
    env_manager=dict(type='base'),
    policy=dict(type='pdqn'),
))
create_config = gym_soccer_pdqn_create_config
