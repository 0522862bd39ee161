"""Predefined per-algorithm example configs for the high-level (bonus) API.

Parity: reference ding/config/example/ (A2C/C51/DDPG/DQN/PG/PPOF/
PPOOffPolicy/SAC/SQL/TD3 per-env cfg modules). Re-scoped to the
self-contained dizoo envs that ship with this build; each entry is a
complete (main_cfg-style) EasyDict selectable by ``get_example_config``.
"""
from ding.utils import EasyDict


def _base_env(env_type: str, import_names: str, stop_value: float, **kwargs):
    return dict(
        env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=stop_value, **kwargs),
        create=dict(
            env=dict(type=env_type, import_names=[import_names]),
            env_manager=dict(type='base'),
        ),
    )


_CARTPOLE = ('cartpole', 'dizoo.classic_control.cartpole.envs.cartpole_env', 195)
_PENDULUM = ('pendulum', 'dizoo.classic_control.pendulum.envs.pendulum_env', -200)
_ATARI = ('atari_lite', 'dizoo.atari.envs.atari_lite_env', 20)

EXAMPLES = {
    'DQN': dict(
        envs={'cartpole': _CARTPOLE, 'atari': _ATARI},
        policy=dict(
            type='dqn', cuda=True, nstep=3, discount_factor=0.99,
            model=dict(encoder_hidden_size_list=[128, 128, 64], dueling=True),
            learn=dict(update_per_collect=10, batch_size=64, learning_rate=1e-3, target_update_freq=100),
            collect=dict(n_sample=64, unroll_len=1),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                       replay_buffer=dict(replay_buffer_size=100000)),
        ),
    ),
    'PPOF': dict(
        envs={'cartpole': _CARTPOLE},
        policy=dict(
            type='ppo', cuda=True, action_space='discrete', recompute_adv=True,
            model=dict(encoder_hidden_size_list=[128, 128, 64]),
            learn=dict(epoch_per_collect=4, batch_size=64, learning_rate=3e-4),
            collect=dict(n_sample=256, discount_factor=0.99, gae_lambda=0.95),
        ),
    ),
    'PPOOffPolicy': dict(
        envs={'cartpole': _CARTPOLE},
        policy=dict(
            type='ppo_offpolicy', cuda=True,
            model=dict(encoder_hidden_size_list=[128, 128, 64]),
            learn=dict(update_per_collect=4, batch_size=64, epoch_per_collect=1, learning_rate=3e-4),
            collect=dict(n_sample=128),
            other=dict(replay_buffer=dict(replay_buffer_size=10000)),
        ),
    ),
    'A2C': dict(
        envs={'cartpole': _CARTPOLE},
        policy=dict(
            type='a2c', cuda=True,
            model=dict(encoder_hidden_size_list=[128, 128, 64]),
            learn=dict(batch_size=64, learning_rate=1e-3),
            collect=dict(n_sample=64, discount_factor=0.99, gae_lambda=0.95),
        ),
    ),
    'PG': dict(
        envs={'cartpole': _CARTPOLE},
        policy=dict(
            type='pg', cuda=True,
            model=dict(),
            learn=dict(batch_size=64, learning_rate=1e-3),
            collect=dict(n_episode=8, discount_factor=0.99),
        ),
    ),
    'C51': dict(
        envs={'cartpole': _CARTPOLE},
        policy=dict(
            type='c51', cuda=True, nstep=3,
            model=dict(v_min=-10, v_max=10, n_atom=51),
            learn=dict(update_per_collect=5, batch_size=64, learning_rate=1e-3),
            collect=dict(n_sample=32),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                       replay_buffer=dict(replay_buffer_size=20000)),
        ),
    ),
    'SQL': dict(
        envs={'cartpole': _CARTPOLE},
        policy=dict(
            type='sql', cuda=True, nstep=1,
            learn=dict(update_per_collect=5, batch_size=64, learning_rate=1e-3, alpha=0.12),
            collect=dict(n_sample=32),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                       replay_buffer=dict(replay_buffer_size=20000)),
        ),
    ),
    'SAC': dict(
        envs={'pendulum': _PENDULUM},
        policy=dict(
            type='sac', cuda=True, random_collect_size=1000,
            model=dict(twin_critic=True, action_space='reparameterization'),
            learn=dict(update_per_collect=1, batch_size=256, auto_alpha=True),
            collect=dict(n_sample=16),
            other=dict(replay_buffer=dict(replay_buffer_size=100000)),
        ),
    ),
    'DDPG': dict(
        envs={'pendulum': _PENDULUM},
        policy=dict(
            type='ddpg', cuda=True, random_collect_size=800,
            model=dict(twin_critic=False, action_space='regression'),
            learn=dict(update_per_collect=2, batch_size=128),
            collect=dict(n_sample=48),
            other=dict(replay_buffer=dict(replay_buffer_size=20000)),
        ),
    ),
    'TD3': dict(
        envs={'pendulum': _PENDULUM},
        policy=dict(
            type='td3', cuda=True, random_collect_size=800,
            model=dict(twin_critic=True, action_space='regression'),
            learn=dict(update_per_collect=2, batch_size=128),
            collect=dict(n_sample=48),
            other=dict(replay_buffer=dict(replay_buffer_size=20000)),
        ),
    ),
}


def get_example_config(algo: str, env: str = None):
    """Return (main_cfg, create_cfg) for a predefined algorithm/env pair."""
    import copy
    entry = EXAMPLES[algo]
    envs = entry['envs']
    env = env or next(iter(envs))
    env_type, import_names, stop_value = envs[env]
    base = _base_env(env_type, import_names, stop_value)
    main = EasyDict(dict(
        exp_name=f'{env}_{algo.lower()}_example',
        env=base['env'],
        policy=copy.deepcopy(entry['policy']),
    ))
    create = EasyDict(dict(
        env=base['create']['env'],
        env_manager=base['create']['env_manager'],
        policy=dict(type=main.policy.pop('type')),
    ))
    return main, create
