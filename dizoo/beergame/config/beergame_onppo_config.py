"""Beer game on-policy PPO (reference
dizoo/beergame/config/beergame_onppo_config.py)."""
from ding.utils import EasyDict

beergame_onppo_config = EasyDict(dict(
    exp_name='beergame_onppo_seed0',
    env=dict(
        role=0,
        weeks=52,
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=-20,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        recompute_adv=True,
        model=dict(obs_shape=5, action_shape=5, action_space='discrete'),
        learn=dict(
            epoch_per_collect=5,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
        ),
        collect=dict(n_sample=1040, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=200, )),
    ),
))
main_config = beergame_onppo_config
beergame_onppo_create_config = EasyDict(dict(
    env=dict(type='beergame', import_names=['dizoo.beergame.envs.beergame_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = beergame_onppo_create_config
