"""Memory-length R2D2 config (recurrent value learning through the fused
LN-LSTM HIP path on MI355X)."""
from ding.utils import EasyDict

main_config = EasyDict(dict(
    exp_name='memory_len_r2d2',
    env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=8, stop_value=0.95,
             memory_length=16),
    policy=dict(
        cuda=True, priority=True, priority_IS_weight=True,
        model=dict(obs_shape=3, action_shape=2, encoder_hidden_size_list=[64, 64], lstm_type='normal'),
        discount_factor=0.997, nstep=3, burnin_step=2, unroll_len=16, learn_unroll_len=14,
        learn=dict(update_per_collect=4, batch_size=32, learning_rate=5e-4, target_update_theta=0.001),
        collect=dict(n_sample=64, unroll_len=16, env_num=4),
        eval=dict(evaluator=dict(eval_freq=200), env_num=4),
        other=dict(eps=dict(type='exp', start=0.95, end=0.05, decay=50000),
                   replay_buffer=dict(replay_buffer_size=50000)),
    ),
))

create_config = EasyDict(dict(
    env=dict(type='memory_len', import_names=['dizoo.memory.envs.memory_len_env']),
    env_manager=dict(type='base'),
    policy=dict(type='r2d2'),
))
