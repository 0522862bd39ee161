"""MPE simple_spread COMA (reference ptz_simple_spread_coma_config.py)."""
from ding.utils import EasyDict

n_agent = 3
n_landmark = n_agent
ptz_simple_spread_coma_config = EasyDict(dict(
    exp_name='ptz_simple_spread_coma_seed0',
    env=dict(
        env_family='mpe',
        env_id='simple_spread_v2',
        n_agent=n_agent,
        n_landmark=n_landmark,
        max_cycles=25,
        agent_obs_only=False,
        continuous_actions=False,
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=0,
    ),
    policy=dict(
        cuda=False,
        model=dict(
            agent_num=n_agent,
            obs_shape=dict(
                agent_state=2 + 2 + n_landmark * 2 + (n_agent - 1) * 2 + (n_agent - 1) * 2,
                global_state=n_agent * 4 + n_landmark * 2 + n_agent * (n_agent - 1) * 2,
            ),
            action_shape=5,
        ),
        learn=dict(
            update_per_collect=1,
            batch_size=32,
            learning_rate=0.0005,
            target_update_theta=0.001,
            discount_factor=0.99,
            td_lambda=0.8,
            value_weight=1.0,
            entropy_weight=0.01,
        ),
        collect=dict(n_sample=600, unroll_len=16, env_num=8),
        eval=dict(env_num=8, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=0.5, end=0.01, decay=100000),
            replay_buffer=dict(replay_buffer_size=10000, ),
        ),
    ),
))
main_config = ptz_simple_spread_coma_config
ptz_simple_spread_coma_create_config = EasyDict(dict(
    env=dict(type='petting_zoo', import_names=['dizoo.petting_zoo.envs.petting_zoo_simple_spread_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='coma'),
))
create_config = ptz_simple_spread_coma_create_config
