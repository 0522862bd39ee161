"""LunarLander discrete SAC (reference lunarlander_discrete_sac_config.py)."""
from ding.utils import EasyDict

lunarlander_disc_sac_config = EasyDict(dict(
    exp_name='lunarlander_discrete_sac_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        env_id='LunarLander-v2',
        n_evaluator_episode=8,
        stop_value=200,
    ),
    policy=dict(
        cuda=False,
        random_collect_size=10000,
        multi_agent=False,
        model=dict(
            obs_shape=8,
            action_shape=4,
            twin_critic=True,
            actor_head_hidden_size=64,
            critic_head_hidden_size=64,
        ),
        learn=dict(
            update_per_collect=1,
            batch_size=64,
            learning_rate_q=5e-4,
            learning_rate_policy=5e-4,
            learning_rate_alpha=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            alpha=0.2,
            auto_alpha=False,
        ),
        collect=dict(n_sample=64, unroll_len=1),
        other=dict(replay_buffer=dict(replay_buffer_size=100000, )),
    ),
))
main_config = lunarlander_disc_sac_config
lunarlander_disc_sac_create_config = EasyDict(dict(
    env=dict(type='lunarlander', import_names=['dizoo.box2d.lunarlander.envs.lunarlander_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='discrete_sac'),
))
create_config = lunarlander_disc_sac_create_config
