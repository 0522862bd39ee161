"""Procgen coinrun PPG (reference coinrun_ppg_config.py: PPO + auxiliary
value-distillation phase)."""
from ding.utils import EasyDict

coinrun_ppg_config = EasyDict(dict(
    exp_name='coinrun_ppg_seed0',
    env=dict(
        env_id='coinrun',
        collector_env_num=16,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=10,
    ),
    policy=dict(
        cuda=True,
        action_space='discrete',
        model=dict(
            obs_shape=[3, 64, 64],
            action_shape=15,
            action_space='discrete',
            encoder_hidden_size_list=[32, 64, 64, 128],
            actor_head_hidden_size=128,
            critic_head_hidden_size=128,
        ),
        learn=dict(
            update_per_collect=1,
            batch_size=512,
            learning_rate=5e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            aux_freq=8,
        ),
        collect=dict(n_sample=4096, unroll_len=1, discount_factor=0.999, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=5000, )),
        other=dict(replay_buffer=dict(
            multi_buffer=True,
            policy=dict(replay_buffer_size=4096),
            value=dict(replay_buffer_size=4096 * 8),
        )),
    ),
))
main_config = coinrun_ppg_config
coinrun_ppg_create_config = EasyDict(dict(
    env=dict(type='procgen', import_names=['dizoo.procgen.envs.procgen_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppg'),
))
create_config = coinrun_ppg_create_config
