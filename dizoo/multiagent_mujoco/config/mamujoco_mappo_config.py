"""Multi-agent MuJoCo HalfCheetah 2x3 MAPPO-continuous (reference
dizoo/multiagent_mujoco/config/halfcheetah_mappo_config.py)."""
from ding.utils import EasyDict

n_agent = 2
mamujoco_mappo_config = EasyDict(dict(
    exp_name='mamujoco_halfcheetah_2x3_mappo_seed0',
    env=dict(
        scenario='2x3',
        base_env='HalfCheetah-v3',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=6000,
    ),
    policy=dict(
        cuda=False,
        multi_agent=True,
        action_space='continuous',
        model=dict(
            action_space='continuous',
            agent_num=n_agent,
            agent_obs_shape=17 + n_agent,
            global_obs_shape=17,
            action_shape=3,
        ),
        learn=dict(
            epoch_per_collect=5,
            batch_size=800,
            learning_rate=5e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95, env_num=8),
        eval=dict(env_num=5, evaluator=dict(eval_freq=200, )),
    ),
))
main_config = mamujoco_mappo_config
mamujoco_mappo_create_config = EasyDict(dict(
    env=dict(type='mamujoco', import_names=['dizoo.multiagent_mujoco.envs.mamujoco_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = mamujoco_mappo_create_config
