"""bsuite memory_len/16 R2D2 (recurrent credit assignment diagnostic;
reference dizoo/bsuite/config/serial/memory_len/memory_len_r2d2_config.py)."""
from ding.utils import EasyDict

memory_len_r2d2_config = EasyDict(dict(
    exp_name='bsuite_memory_len_r2d2_seed0',
    env=dict(
        env_id='memory_len/16',
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=10,
        stop_value=0.95,
    ),
    policy=dict(
        cuda=False,
        priority=True,
        priority_IS_weight=True,
        model=dict(obs_shape=3, action_shape=2, encoder_hidden_size_list=[64, 64], lstm_type='normal'),
        discount_factor=0.997,
        nstep=3,
        burnin_step=2,
        unroll_len=18,
        learn_unroll_len=16,
        learn=dict(update_per_collect=4, batch_size=32, learning_rate=5e-4, target_update_theta=0.001),
        collect=dict(n_sample=64, unroll_len=18, env_num=8),
        eval=dict(env_num=5, evaluator=dict(eval_freq=200, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.05, decay=50000),
            replay_buffer=dict(type='advanced', replay_buffer_size=50000),
        ),
    ),
))
main_config = memory_len_r2d2_config
memory_len_r2d2_create_config = EasyDict(dict(
    env=dict(type='bsuite', import_names=['dizoo.bsuite.envs.bsuite_env']),
    env_manager=dict(type='base'),
    policy=dict(type='r2d2'),
))
create_config = memory_len_r2d2_create_config
