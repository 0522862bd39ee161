"""gym-hybrid Moving DDPG (PADDPG-style hybrid QAC; reference
gym_hybrid_ddpg_config.py, stop_value 1.8)."""
from ding.utils import EasyDict

gym_hybrid_ddpg_config = EasyDict(dict(
    exp_name='gym_hybrid_ddpg_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        env_id='Moving-v0',
        act_scale=True,
        stop_value=1.8,
    ),
    policy=dict(
        cuda=False,
        random_collect_size=0,
        action_space='hybrid',
        model=dict(
            obs_shape=10,
            action_shape=dict(
                action_type_shape=3,
                action_args_shape=2,
            ),
            action_space='hybrid',
            twin_critic=False,
        ),
        learn=dict(
            update_per_collect=10,
            batch_size=32,
            learning_rate_actor=3e-4,
            learning_rate_critic=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            actor_update_freq=1,
            noise=False,
        ),
        collect=dict(n_sample=32, unroll_len=1, noise_sigma=0.1),
        other=dict(
            eps=dict(type='exp', start=1., end=0.1, decay=100000),
            replay_buffer=dict(replay_buffer_size=100000, ),
        ),
    ),
))
main_config = gym_hybrid_ddpg_config
gym_hybrid_ddpg_create_config = EasyDict(dict(
    env=dict(type='gym_hybrid', import_names=['dizoo.gym_hybrid.envs.moving_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ddpg'),
))
create_config = gym_hybrid_ddpg_create_config
