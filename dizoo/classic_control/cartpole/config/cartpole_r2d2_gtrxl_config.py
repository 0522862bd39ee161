"""CartPole R2D2-GTrXL (transformer memory instead of LSTM burn-in;
reference cartpole_r2d2_gtrxl_config.py)."""
from ding.utils import EasyDict

cartpole_r2d2_gtrxl_config = EasyDict(dict(
    exp_name='cartpole_r2d2_gtrxl_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=195,
    ),
    policy=dict(
        cuda=False,
        priority=True,
        priority_IS_weight=True,
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[128, 128, 64],
                   hidden_size=64, att_head_num=2, att_layer_num=2, memory_len=4),
        discount_factor=0.997,
        nstep=3,
        burnin_step=0,
        unroll_len=16,
        learn_unroll_len=14,
        seq_len=16,
        learn=dict(update_per_collect=4, batch_size=32, learning_rate=5e-4, target_update_theta=0.001,
                   value_rescale=True, init_memory='zero'),
        collect=dict(n_sample=64, unroll_len=16, env_num=8),
        eval=dict(env_num=5, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.05, decay=10000),
            replay_buffer=dict(type='advanced', replay_buffer_size=10000),
        ),
    ),
))
main_config = cartpole_r2d2_gtrxl_config
cartpole_r2d2_gtrxl_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='r2d2_gtrxl'),
))
create_config = cartpole_r2d2_gtrxl_create_config
