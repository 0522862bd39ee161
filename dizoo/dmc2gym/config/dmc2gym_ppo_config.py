"""dmc2gym cheetah-run on-policy PPO continuous (reference
dizoo/dmc2gym/config/dmc2gym_ppo_config.py)."""
from ding.utils import EasyDict

dmc2gym_ppo_config = EasyDict(dict(
    exp_name='dmc2gym_cheetah_run_ppo_seed0',
    env=dict(
        env_id='dmc2gym-cheetah-run',
        domain_name='cheetah',
        task_name='run',
        from_pixels=False,
        frame_skip=1,
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=750,
    ),
    policy=dict(
        cuda=True,
        action_space='continuous',
        recompute_adv=True,
        model=dict(
            obs_shape=17,
            action_shape=6,
            action_space='continuous',
        ),
        learn=dict(
            epoch_per_collect=10,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = dmc2gym_ppo_config
dmc2gym_ppo_create_config = EasyDict(dict(
    env=dict(type='dmc2gym', import_names=['dizoo.dmc2gym.envs.dmc2gym_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
))
create_config = dmc2gym_ppo_create_config
