"""Factory for the MuJoCo per-env config families (reference ships one file
per (env, algorithm): dizoo/mujoco/config/<env>_<algo>_config.py). All run
on the mujoco-lite linear-dynamics env (no MuJoCo binary in this image) with
the reference's obs/action dims per env.
"""
import copy

from ding.utils import EasyDict

ENVS = {
    'hopper': ('Hopper-v3', 11, 3, 6000),
    'halfcheetah': ('HalfCheetah-v3', 17, 6, 12000),
    'walker2d': ('Walker2d-v3', 17, 6, 6000),
    'ant': ('Ant-v3', 111, 8, 6000),
    'humanoid': ('Humanoid-v3', 376, 17, 8000),
}


def _env_block(env: str, collector_env_num: int = 1) -> dict:
    env_id, _, _, stop = ENVS[env]
    return dict(
        env_id=env_id,
        collector_env_num=collector_env_num,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=stop,
    )


def _policy_block(algo: str, obs: int, act: int):
    if algo == 'sac':
        pol = dict(
            cuda=True,
            priority=False,
            random_collect_size=10000,
            model=dict(
                obs_shape=obs, action_shape=act, action_space='reparameterization', twin_critic=True,
                actor_head_hidden_size=256, critic_head_hidden_size=256,
            ),
            learn=dict(
                update_per_collect=1, batch_size=256, learning_rate_q=1e-3, learning_rate_policy=1e-3,
                learning_rate_alpha=3e-4, target_theta=0.005, discount_factor=0.99, auto_alpha=False,
            ),
            collect=dict(n_sample=1, unroll_len=1),
            eval=dict(evaluator=dict(eval_freq=5000, )),
            other=dict(replay_buffer=dict(replay_buffer_size=1000000, )),
        )
        return pol, 'sac', 'serial'
    if algo in ('td3', 'ddpg'):
        pol = dict(
            cuda=True,
            priority=False,
            random_collect_size=25000 if algo == 'td3' else 10000,
            model=dict(
                obs_shape=obs, action_shape=act, action_space='regression', twin_critic=(algo == 'td3'),
                actor_head_hidden_size=256, critic_head_hidden_size=256,
            ),
            learn=dict(
                update_per_collect=1, batch_size=256, learning_rate_actor=1e-3, learning_rate_critic=1e-3,
                target_theta=0.005, discount_factor=0.99,
                actor_update_freq=2 if algo == 'td3' else 1,
                noise=(algo == 'td3'), noise_sigma=0.2, noise_range=dict(min=-0.5, max=0.5),
            ),
            collect=dict(n_sample=1, unroll_len=1, noise_sigma=0.1),
            eval=dict(evaluator=dict(eval_freq=5000, )),
            other=dict(replay_buffer=dict(replay_buffer_size=1000000, )),
        )
        return pol, algo, 'serial'
    if algo == 'd4pg':
        pol = dict(
            cuda=True,
            priority=True,
            nstep=5,
            random_collect_size=10000,
            model=dict(
                obs_shape=obs, action_shape=act, action_space='regression',
                actor_head_hidden_size=256, critic_head_hidden_size=256,
                v_min=-100, v_max=100, n_atom=51,
            ),
            learn=dict(
                update_per_collect=1, batch_size=256, learning_rate_actor=1e-3, learning_rate_critic=1e-3,
                target_theta=0.005, discount_factor=0.99,
            ),
            collect=dict(n_sample=1, unroll_len=1, noise_sigma=0.1),
            eval=dict(evaluator=dict(eval_freq=5000, )),
            other=dict(replay_buffer=dict(replay_buffer_size=1000000, )),
        )
        return pol, 'd4pg', 'serial'
    if algo == 'onppo':
        pol = dict(
            cuda=True,
            recompute_adv=True,
            action_space='continuous',
            model=dict(
                obs_shape=obs, action_shape=act, action_space='continuous',
                encoder_hidden_size_list=[128, 128],
                actor_head_hidden_size=128, critic_head_hidden_size=128,
            ),
            learn=dict(
                epoch_per_collect=10, update_per_collect=1, batch_size=320, learning_rate=3e-4,
                value_weight=0.5, entropy_weight=0.001, clip_ratio=0.2, adv_norm=True, value_norm=True,
            ),
            collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
            eval=dict(evaluator=dict(eval_freq=5000, )),
        )
        return pol, 'ppo', 'onpolicy'
    if algo == 'bdq':
        # no random warm-up: the env's random actions are continuous, BDQ's
        # replay expects discretized branch indices from its own collect
        pol = dict(
            cuda=True,
            priority=False,
            model=dict(
                obs_shape=obs, num_branches=act, action_bins_per_branch=4,
                encoder_hidden_size_list=[256, 256, 128],
            ),
            nstep=3,
            discount_factor=0.99,
            learn=dict(update_per_collect=10, batch_size=512, learning_rate=3e-4, target_update_freq=500),
            collect=dict(n_sample=256, ),
            eval=dict(evaluator=dict(eval_freq=5000, )),
            other=dict(
                eps=dict(type='exp', start=0.95, end=0.05, decay=10000),
                replay_buffer=dict(replay_buffer_size=1000000, ),
            ),
        )
        return pol, 'bdq', 'serial'
    raise KeyError(f"unknown mujoco algo: {algo}")


def build_mujoco_config(env: str, algo: str):
    env_id, obs, act, _ = ENVS[env]
    policy, policy_type, pipeline = _policy_block(algo, obs, act)
    main_config = EasyDict(dict(
        exp_name=f'{env}_{algo}_seed0',
        env=_env_block(env),
        policy=policy,
    ))
    create_config = EasyDict(dict(
        env=dict(type='mujoco_lite', import_names=['dizoo.mujoco.envs.mujoco_lite_env']),
        env_manager=dict(type='base'),
        policy=dict(type=policy_type),
    ))
    main_config._pipeline = pipeline
    return main_config, create_config
