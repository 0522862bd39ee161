"""Pong DQN (BASELINE row: eval return >= 18 on real ALE; atari-lite here;
reference dizoo/atari/config/serial/pong/pong_dqn_config.py)."""
from ding.utils import EasyDict

pong_dqn_config = EasyDict(dict(
    exp_name='pong_dqn_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=18,
        env_id='PongNoFrameskip-v4',
        frame_stack=4,
    ),
    policy=dict(
        cuda=True,
        priority=False,
        model=dict(
            obs_shape=[4, 84, 84],
            action_shape=6,
            encoder_hidden_size_list=[128, 128, 512],
        ),
        nstep=3,
        discount_factor=0.99,
        learn=dict(
            update_per_collect=10,
            batch_size=32,
            learning_rate=1e-4,
            target_update_freq=500,
        ),
        collect=dict(n_sample=96, ),
        eval=dict(evaluator=dict(eval_freq=4000, )),
        other=dict(
            eps=dict(type='exp', start=1., end=0.05, decay=250000),
            replay_buffer=dict(replay_buffer_size=400000, ),
        ),
    ),
))
main_config = pong_dqn_config
pong_dqn_create_config = EasyDict(dict(
    env=dict(type='atari_lite', import_names=['dizoo.atari.envs.atari_lite_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='dqn'),
))
create_config = pong_dqn_create_config
