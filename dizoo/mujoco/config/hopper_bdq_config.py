"""Hopper BDQ: branching dueling Q over a discretized action space
(reference dizoo/mujoco/config/hopper_bdq_config.py)."""
from ding.utils import EasyDict

hopper_bdq_config = EasyDict(dict(
    exp_name='hopper_bdq_seed0',
    env=dict(
        env_id='Hopper-v3',
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=6000,
    ),
    policy=dict(
        cuda=True,
        nstep=3,
        discount_factor=0.99,
        model=dict(
            obs_shape=11,
            num_branches=3,       # one branch per action dim
            action_bins_per_branch=4,
            encoder_hidden_size_list=[256, 256, 128],
        ),
        learn=dict(update_per_collect=10, batch_size=512, learning_rate=3e-4, target_update_freq=500),
        collect=dict(n_sample=256, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=100000),
            replay_buffer=dict(replay_buffer_size=1000000, ),
        ),
    ),
))
main_config = hopper_bdq_config
hopper_bdq_create_config = EasyDict(dict(
    env=dict(type='mujoco_lite', import_names=['dizoo.mujoco.envs.mujoco_lite_env']),
    env_manager=dict(type='base'),
    policy=dict(type='bdq'),
))
create_config = hopper_bdq_create_config
