"""MPE simple_spread COLLAQ (reference
dizoo/petting_zoo/config/ptz_simple_spread_collaq_config.py; n_agent=3,
obs 18 / global 30 / 5 actions, stop_value 0)."""
from ding.utils import EasyDict

n_agent = 3
n_landmark = n_agent
ptz_simple_spread_collaq_config = EasyDict(dict(
    exp_name='ptz_simple_spread_collaq_seed0',
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
            obs_shape=2 + 2 + n_landmark * 2 + (n_agent - 1) * 2 + (n_agent - 1) * 2,
            global_obs_shape=n_agent * 4 + n_landmark * 2 + n_agent * (n_agent - 1) * 2,
            action_shape=5,
            hidden_size_list=[128, 128, 64],
            mixer=True,
        ),
        learn=dict(
            update_per_collect=100,
            batch_size=32,
            learning_rate=0.0005,
            target_update_theta=0.001,
            discount_factor=0.99,
        ),
        collect=dict(n_sample=600, unroll_len=16, env_num=8),
        eval=dict(env_num=8, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=1.0, end=0.05, decay=100000),
            replay_buffer=dict(replay_buffer_size=15000, ),
        ),
    ),
))
main_config = ptz_simple_spread_collaq_config
ptz_simple_spread_collaq_create_config = EasyDict(dict(
    env=dict(type='petting_zoo', import_names=['dizoo.petting_zoo.envs.petting_zoo_simple_spread_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='collaq'),
))
create_config = ptz_simple_spread_collaq_create_config
