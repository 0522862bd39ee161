"""MPE simple_spread HAPPO (reference ptz_simple_spread_happo_config.py)."""
from ding.utils import EasyDict

n_agent = 3
n_landmark = n_agent
ptz_simple_spread_happo_config = EasyDict(dict(
    exp_name='ptz_simple_spread_happo_seed0',
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
        agent_num=n_agent,
        action_space='discrete',
        model=dict(
            action_space='discrete',
            agent_num=n_agent,
            agent_obs_shape=2 + 2 + n_landmark * 2 + (n_agent - 1) * 2 + (n_agent - 1) * 2,
            global_obs_shape=n_agent * 4 + n_landmark * 2 + n_agent * (n_agent - 1) * 2,
            action_shape=5,
        ),
        learn=dict(
            epoch_per_collect=5,
            batch_size=800,
            learning_rate=5e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95, env_num=8),
        eval=dict(env_num=8, evaluator=dict(eval_freq=50, )),
    ),
))
main_config = ptz_simple_spread_happo_config
ptz_simple_spread_happo_create_config = EasyDict(dict(
    env=dict(type='petting_zoo', import_names=['dizoo.petting_zoo.envs.petting_zoo_simple_spread_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='happo'),
))
create_config = ptz_simple_spread_happo_create_config
