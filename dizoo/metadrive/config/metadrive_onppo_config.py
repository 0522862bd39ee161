"""MetaDrive-lite macro driving, on-policy PPO continuous (reference
dizoo/metadrive/config/metadrive_onppo_config.py)."""
from ding.utils import EasyDict

metadrive_onppo_config = EasyDict(dict(
    exp_name='metadrive_onppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=50,
    ),
    policy=dict(
        cuda=True,
        action_space='continuous',
        recompute_adv=True,
        model=dict(obs_shape=17, action_shape=2, action_space='continuous'),
        learn=dict(epoch_per_collect=10, batch_size=320, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.001, clip_ratio=0.2, adv_norm=True, value_norm=True),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = metadrive_onppo_config
metadrive_onppo_create_config = EasyDict(dict(
    env=dict(type='metadrive', import_names=['dizoo.metadrive.envs.metadrive_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = metadrive_onppo_create_config
