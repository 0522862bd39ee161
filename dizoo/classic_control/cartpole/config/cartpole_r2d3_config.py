"""CartPole R2D3 (recurrent DQfD: demo buffer mixed per batch; reference
cartpole_r2d3_r2d2expert_config.py; run with serial_pipeline_r2d3)."""
from ding.utils import EasyDict

cartpole_r2d3_config = EasyDict(dict(
    exp_name='cartpole_r2d3_seed0',
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
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[128, 128, 64]),
        discount_factor=0.997,
        nstep=5,
        burnin_step=2,
        unroll_len=16,
        learn_unroll_len=14,
        lambda1=1.0,
        lambda2=1.0,
        margin_function=0.8,
        learn=dict(update_per_collect=4, batch_size=32, learning_rate=5e-4, target_update_theta=0.001),
        collect=dict(n_sample=64, unroll_len=16, env_num=8, pho=0.25),
        eval=dict(env_num=5, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.05, decay=10000),
            replay_buffer=dict(type='advanced', replay_buffer_size=10000),
        ),
    ),
))
main_config = cartpole_r2d3_config
cartpole_r2d3_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='r2d3'),
))
create_config = cartpole_r2d3_create_config
