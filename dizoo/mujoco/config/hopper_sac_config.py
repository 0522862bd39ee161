"""Hopper SAC + PER workload (BASELINE.json config #4 analog; reference
dizoo/mujoco/config/hopper_sac_config.py, stop_value 6000 on real MuJoCo)."""
from ding.utils import EasyDict

hopper_sac_config = dict(
    exp_name='hopper_sac_seed0',
    env=dict(
        env_id='Hopper-v3',
        collector_env_num=1,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=6000,
    ),
    policy=dict(
        cuda=True,
        priority=True,
        priority_IS_weight=True,
        random_collect_size=10000,
        model=dict(
            obs_shape=11,
            action_shape=3,
            action_space='reparameterization',
            twin_critic=True,
            actor_head_hidden_size=256,
            critic_head_hidden_size=256,
        ),
        learn=dict(
            update_per_collect=1,
            batch_size=256,
            learning_rate_q=1e-3,
            learning_rate_policy=1e-3,
            learning_rate_alpha=3e-4,
            ignore_done=False,
            target_theta=0.005,
            discount_factor=0.99,
            alpha=0.2,
            auto_alpha=False,
        ),
        collect=dict(n_sample=1, unroll_len=1, ),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(replay_buffer=dict(type='advanced', replay_buffer_size=1000000, ), ),
    ),
)
main_config = EasyDict(hopper_sac_config)
hopper_sac_create_config = dict(
    env=dict(type='mujoco_lite', import_names=['dizoo.mujoco.envs.mujoco_lite_env']),
    env_manager=dict(type='base'),
    policy=dict(type='sac'),
)
create_config = EasyDict(hopper_sac_create_config)
