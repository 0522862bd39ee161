"""Ising model mean-field cooperation with QMIX (reference
dizoo/ising_env/config/ising_mf_q_config.py adapted to the QMIX lane)."""
from ding.utils import EasyDict

n_agent = 10
ising_mf_qmix_config = EasyDict(dict(
    exp_name='ising_mf_qmix_seed0',
    env=dict(
        num_agents=n_agent,
        agent_view_sight=2,
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=45,
    ),
    policy=dict(
        cuda=False,
        model=dict(
            agent_num=n_agent,
            obs_shape=5,
            global_obs_shape=n_agent,
            action_shape=2,
            hidden_size_list=[64, 64],
            mixer=True,
        ),
        learn=dict(
            update_per_collect=20,
            batch_size=32,
            learning_rate=5e-4,
            target_update_theta=0.001,
            discount_factor=0.99,
        ),
        collect=dict(n_sample=400, unroll_len=10, env_num=8),
        eval=dict(env_num=5, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=1.0, end=0.05, decay=50000),
            replay_buffer=dict(replay_buffer_size=5000),
        ),
    ),
))
main_config = ising_mf_qmix_config
ising_mf_qmix_create_config = EasyDict(dict(
    env=dict(type='ising_model', import_names=['dizoo.ising_env.envs.ising_model_env']),
    env_manager=dict(type='base'),
    policy=dict(type='qmix'),
))
create_config = ising_mf_qmix_create_config
