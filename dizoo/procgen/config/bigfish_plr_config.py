"""Procgen bigfish PPO + Prioritized Level Replay (reference
dizoo/procgen/config/bigfish_plr_config.py, stop_value 40). Run with
ding.entry.serial_pipeline_plr."""
from ding.utils import EasyDict

bigfish_plr_config = EasyDict(dict(
    exp_name='bigfish_plr_seed0',
    env=dict(
        env_id='bigfish',
        collector_env_num=16,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=40,
    ),
    level_replay=dict(
        num_seeds=128,
        strategy='policy_entropy',
        score_transform='rank',
        temperature=0.1,
    ),
    policy=dict(
        cuda=True,
        action_space='discrete',
        recompute_adv=True,
        model=dict(
            obs_shape=[3, 64, 64],
            action_shape=15,
            action_space='discrete',
            encoder_hidden_size_list=[32, 64, 64, 128],
            actor_head_hidden_size=128,
            critic_head_hidden_size=128,
        ),
        learn=dict(
            epoch_per_collect=3,
            batch_size=512,
            learning_rate=5e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
            ignore_done=False,
        ),
        collect=dict(n_sample=4096, unroll_len=1, discount_factor=0.999, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=5000, )),
    ),
))
main_config = bigfish_plr_config
bigfish_plr_create_config = EasyDict(dict(
    env=dict(type='procgen', import_names=['dizoo.procgen.envs.procgen_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
))
create_config = bigfish_plr_create_config
