"""Overcooked cramped-room QMIX (reference
dizoo/overcooked/config/overcooked_demo_ppo_config.py lane, on the QMIX
cooperative stack)."""
from ding.utils import EasyDict

overcooked_qmix_config = EasyDict(dict(
    exp_name='overcooked_qmix_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=60,
    ),
    policy=dict(
        cuda=False,
        model=dict(
            agent_num=2,
            obs_shape=10,
            global_obs_shape=8,
            action_shape=6,
            hidden_size_list=[64, 64],
            mixer=True,
        ),
        learn=dict(update_per_collect=20, batch_size=32, learning_rate=5e-4,
                   target_update_theta=0.001, discount_factor=0.99),
        collect=dict(n_sample=600, unroll_len=16, env_num=8),
        eval=dict(env_num=5, evaluator=dict(eval_freq=200, )),
        other=dict(
            eps=dict(type='exp', start=1.0, end=0.05, decay=100000),
            replay_buffer=dict(replay_buffer_size=5000),
        ),
    ),
))
main_config = overcooked_qmix_config
overcooked_qmix_create_config = EasyDict(dict(
    env=dict(type='overcooked', import_names=['dizoo.overcooked.envs.overcooked_env']),
    env_manager=dict(type='base'),
    policy=dict(type='qmix'),
))
create_config = overcooked_qmix_create_config
