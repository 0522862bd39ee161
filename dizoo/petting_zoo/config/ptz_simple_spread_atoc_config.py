"""MPE simple_spread ATOC (communicating MADDPG with attention gates;
reference ptz_simple_spread_atoc_config.py — continuous MPE actions)."""
from ding.utils import EasyDict

n_agent = 3
n_landmark = n_agent
obs_dim = 2 + 2 + n_landmark * 2 + (n_agent - 1) * 2 + (n_agent - 1) * 2
ptz_simple_spread_atoc_config = EasyDict(dict(
    exp_name='ptz_simple_spread_atoc_seed0',
    env=dict(
        env_family='mpe',
        env_id='simple_spread_v2',
        n_agent=n_agent,
        n_landmark=n_landmark,
        max_cycles=25,
        agent_obs_only=True,
        continuous_actions=True,
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=0,
    ),
    policy=dict(
        cuda=False,
        model=dict(
            obs_shape=obs_dim,
            action_shape=2,
            n_agent=n_agent,
            thought_size=16,
            agent_per_group=2,
            communication=True,
        ),
        learn=dict(
            update_per_collect=5,
            batch_size=32,
            learning_rate_actor=1e-3,
            learning_rate_critic=1e-3,
            target_theta=0.005,
            discount_factor=0.99,
            communication=True,
            actor_update_freq=1,
            noise=True,
            noise_sigma=0.15,
            noise_range=dict(min=-0.5, max=0.5),
        ),
        collect=dict(n_sample=500, unroll_len=1, noise_sigma=0.4),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=100000, )),
    ),
))
main_config = ptz_simple_spread_atoc_config
ptz_simple_spread_atoc_create_config = EasyDict(dict(
    env=dict(type='petting_zoo', import_names=['dizoo.petting_zoo.envs.petting_zoo_simple_spread_env']),
    env_manager=dict(type='base'),
    policy=dict(type='atoc'),
))
create_config = ptz_simple_spread_atoc_create_config
