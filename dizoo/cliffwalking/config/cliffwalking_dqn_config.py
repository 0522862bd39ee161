"""CliffWalking DQN (reference dizoo/cliffwalking/config/
cliffwalking_dqn_config.py, stop_value -13 = optimal path)."""
from ding.utils import EasyDict

cliffwalking_dqn_config = EasyDict(dict(
    exp_name='cliffwalking_dqn_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=-13,
        max_step=300,
    ),
    policy=dict(
        cuda=False,
        model=dict(
            obs_shape=48,
            action_shape=4,
            encoder_hidden_size_list=[512, 64],
        ),
        discount_factor=0.98,
        nstep=1,
        learn=dict(update_per_collect=10, batch_size=128, learning_rate=1e-4, target_update_freq=100),
        collect=dict(n_sample=64),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='linear', start=1.0, end=0.05, decay=3000000),
            replay_buffer=dict(replay_buffer_size=100000, ),
        ),
    ),
))
main_config = cliffwalking_dqn_config
cliffwalking_dqn_create_config = EasyDict(dict(
    env=dict(type='cliffwalking', import_names=['dizoo.cliffwalking.envs.cliffwalking_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='dqn'),
))
create_config = cliffwalking_dqn_create_config
