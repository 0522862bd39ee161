"""Tuned C51 preset for PongNoFrameskip-v4 (reference
ding/config/example/C51/gym_pongnoframeskip_v4.py)."""
from ding.utils import EasyDict

cfg = EasyDict(dict(
    exp_name='PongNoFrameskip-v4-C51',
    seed=0,
    env=dict(
        type='atari_lite',
        import_names=['dizoo.atari.envs.atari_lite_env'],
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=20,
        frame_stack=4,
    ),
    policy=dict(
        cuda=True,
        discount_factor=0.99,
        nstep=3,
        model=dict(obs_shape=[4, 84, 84], action_shape=6, encoder_hidden_size_list=[128, 128, 64],
                   v_min=-30, v_max=30, n_atom=51),
        learn=dict(update_per_collect=10, batch_size=64, learning_rate=1e-3, target_update_freq=100),
        collect=dict(n_sample=64, unroll_len=1),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=50000),
            replay_buffer=dict(replay_buffer_size=100000),
        ),
    ),
))
