"""SpaceInvaders IMPALA workload config (BASELINE.json config #3):
unroll_len=32, batch=128, update_per_collect=2, 8 envs. Mirrors the
reference dizoo/atari/config/serial/spaceinvaders/
spaceinvaders_impala_config.py:17-45 on atari-lite."""
from ding.utils import EasyDict

spaceinvaders_impala_config = dict(
    exp_name='spaceinvaders_impala_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=int(1e10),
        env_id='SpaceInvadersNoFrameskip-v4',
        frame_stack=4,
    ),
    policy=dict(
        cuda=True,
        unroll_len=32,
        random_collect_size=0,
        model=dict(
            obs_shape=[4, 84, 84],
            action_shape=6,
            encoder_hidden_size_list=[128, 128, 256],
            critic_head_hidden_size=256,
            critic_head_layer_num=3,
            actor_head_hidden_size=256,
            actor_head_layer_num=3,
        ),
        learn=dict(
            update_per_collect=2,
            batch_size=128,
            grad_clip_type='clip_norm',
            clip_value=5,
            learning_rate=0.0006,
            value_weight=0.5,
            entropy_weight=0.01,
            discount_factor=0.99,
            lambda_=0.95,
            rho_clip_ratio=1.0,
            c_clip_ratio=1.0,
            rho_pg_clip_ratio=1.0,
        ),
        collect=dict(n_sample=16, ),
        eval=dict(evaluator=dict(eval_freq=5000, )),
        other=dict(replay_buffer=dict(replay_buffer_size=1000, ), ),
    ),
)
main_config = EasyDict(spaceinvaders_impala_config)
spaceinvaders_impala_create_config = dict(
    env=dict(type='atari_lite', import_names=['dizoo.atari.envs.atari_lite_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='impala'),
)
create_config = EasyDict(spaceinvaders_impala_create_config)
