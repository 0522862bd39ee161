"""Pendulum STEVE-SAC + DDPPO gradient-through-dynamics ensemble (reference
dizoo mbrl config family; run with ding.entry.serial_pipeline_dream)."""
from ding.utils import EasyDict

pendulum_stevesac_ddppo_config = EasyDict(dict(
    exp_name='pendulum_stevesac_ddppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=-250,
        act_scale=True,
    ),
    policy=dict(
        cuda=True,
        random_collect_size=1000,
        model=dict(obs_shape=3, action_shape=1, action_space='reparameterization', twin_critic=True),
        learn=dict(
            ensemble_size=5,
            update_per_collect=2,
            batch_size=128,
            learning_rate_q=1e-3,
            learning_rate_policy=1e-3,
            learning_rate_alpha=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            auto_alpha=False, alpha=0.2,
        ),
        collect=dict(n_sample=10, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=100, )),
        other=dict(replay_buffer=dict(replay_buffer_size=100000)),
    ),
    world_model=dict(
        type='ddppo',
        import_names=['ding.world_model.ddppo'],
        train_freq=250,
        eval_freq=250,
        cuda=True,
        model=dict(state_size=3, action_size=1, hidden_size=256, ensemble_size=5, elite_size=3,
                   batch_size=256),
        other=dict(
            real_ratio=0.05,
            rollout_batch_size=10000,
            imagination_buffer=dict(replay_buffer_size=600000),
        ),
        rollout_length_scheduler=dict(rollout_start_step=2000, rollout_end_step=15000,
                                      rollout_length_min=1, rollout_length_max=1),
    ),
))
main_config = pendulum_stevesac_ddppo_config
pendulum_stevesac_ddppo_create_config = EasyDict(dict(
    env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
    env_manager=dict(type='base'),
    policy=dict(type='stevesac'),
    world_model=dict(type='ddppo', import_names=['ding.world_model.ddppo']),
))
create_config = pendulum_stevesac_ddppo_create_config
