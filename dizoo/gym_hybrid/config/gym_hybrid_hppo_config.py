"""gym-hybrid Moving HPPO: on-policy PPO with hybrid action space (reference
gym_hybrid_hppo_config.py — policy type 'ppo', action_space='hybrid')."""
from ding.utils import EasyDict

gym_hybrid_hppo_config = EasyDict(dict(
    exp_name='gym_hybrid_hppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        env_id='Moving-v0',
        act_scale=True,
        stop_value=1.8,
    ),
    policy=dict(
        cuda=False,
        action_space='hybrid',
        recompute_adv=True,
        model=dict(
            obs_shape=10,
            action_shape=dict(
                action_type_shape=3,
                action_args_shape=2,
            ),
            action_space='hybrid',
            encoder_hidden_size_list=[256, 128, 64],
            sigma_type='fixed',
            fixed_sigma_value=0.3,
            bound_type='tanh',
        ),
        learn=dict(
            epoch_per_collect=10,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.03,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
    ),
))
main_config = gym_hybrid_hppo_config
gym_hybrid_hppo_create_config = EasyDict(dict(
    env=dict(type='gym_hybrid', import_names=['dizoo.gym_hybrid.envs.moving_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
))
create_config = gym_hybrid_hppo_create_config
