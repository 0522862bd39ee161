"""Pong PPO workload config (BASELINE.json config #2): n_sample=3200,
batch=320, epoch_per_collect=10, 8 collector envs. Mirrors the reference
dizoo/atari/config/serial/pong/pong_ppo_config.py:5-56 on the atari-lite
synthetic env (no ALE offline)."""
from ding.utils import EasyDict

pong_ppo_config = dict(
    exp_name='pong_ppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=20,
        env_id='PongNoFrameskip-v4',
        frame_stack=4,
    ),
    policy=dict(
        cuda=True,
        recompute_adv=True,
        action_space='discrete',
        model=dict(
            obs_shape=[4, 84, 84],
            action_shape=6,
            action_space='discrete',
            encoder_hidden_size_list=[64, 64, 128],
            actor_head_hidden_size=128,
            critic_head_hidden_size=128,
        ),
        learn=dict(
            epoch_per_collect=10,
            update_per_collect=1,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
            ignore_done=False,
            grad_clip_type='clip_norm',
            grad_clip_value=0.5,
        ),
        collect=dict(
            n_sample=3200,
            unroll_len=1,
            discount_factor=0.99,
            gae_lambda=0.95,
        ),
        eval=dict(evaluator=dict(eval_freq=5000, )),
    ),
)
main_config = EasyDict(pong_ppo_config)
pong_ppo_create_config = dict(
    env=dict(type='atari_lite', import_names=['dizoo.atari.envs.atari_lite_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='ppo'),
)
create_config = EasyDict(pong_ppo_create_config)
