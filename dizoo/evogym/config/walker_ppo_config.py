"""EvoGym Walker-v0 PPO (reference dizoo/evogym/config/walker_ppo_config.py;
the soft-body simulator is unavailable offline so the mujoco-lite smooth
dynamics stand in at the evogym observation/action shapes: obs 58, act 10)."""
from ding.utils import EasyDict

walker_ppo_config = EasyDict(dict(
    exp_name='evogym_walker_ppo_seed0',
    env=dict(
        env_id='Walker-v0',
        obs_dim=58,
        act_dim=10,
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=10,
    ),
    policy=dict(
        cuda=True,
        action_space='continuous',
        recompute_adv=True,
        model=dict(obs_shape=58, action_shape=10, action_space='continuous'),
        learn=dict(epoch_per_collect=10, batch_size=320, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.001, clip_ratio=0.2, adv_norm=True, value_norm=True),
        collect=dict(n_sample=2048, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = walker_ppo_config
walker_ppo_create_config = EasyDict(dict(
    env=dict(type='mujoco_lite', import_names=['dizoo.mujoco.envs.mujoco_lite_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = walker_ppo_create_config
