"""dmc2gym cartpole-swingup SAC from pixels (reference
dizoo/dmc2gym/config/dmc2gym_sac_pixel_config.py: 3x84x84, frame_skip 4)."""
from ding.utils import EasyDict

dmc2gym_sac_pixel_config = EasyDict(dict(
    exp_name='dmc2gym_cartpole_swingup_sac_pixel_seed0',
    env=dict(
        env_id='dmc2gym-cartpole-swingup',
        domain_name='cartpole',
        task_name='swingup',
        from_pixels=True,
        channels_first=True,
        frame_skip=4,
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=180,
    ),
    policy=dict(
        cuda=True,
        random_collect_size=5000,
        model=dict(
            obs_shape=[3, 84, 84],
            action_shape=1,
            encoder_hidden_size_list=[32, 32, 64],
            action_space='reparameterization',
            twin_critic=True,
            actor_head_hidden_size=256,
            critic_head_hidden_size=256,
        ),
        learn=dict(
            update_per_collect=1,
            batch_size=128,
            learning_rate_q=1e-3,
            learning_rate_policy=1e-3,
            learning_rate_alpha=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            auto_alpha=True,
        ),
        collect=dict(n_sample=1, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(replay_buffer=dict(replay_buffer_size=100000, )),
    ),
))
main_config = dmc2gym_sac_pixel_config
dmc2gym_sac_pixel_create_config = EasyDict(dict(
    env=dict(type='dmc2gym', import_names=['dizoo.dmc2gym.envs.dmc2gym_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='sac'),
))
create_config = dmc2gym_sac_pixel_create_config
