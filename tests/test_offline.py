"""Offline-RL smoke: synthetic npz dataset -> BC/CQL/TD3BC/IQL/EDAC/discrete-CQL."""
import numpy as np
import pytest
import torch

from ding.entry import serial_pipeline_offline
from ding.utils import EasyDict


@pytest.fixture(scope="module")
def pendulum_npz(tmp_path_factory):
    path = str(tmp_path_factory.mktemp("data") / "pendulum.npz")
    n = 256
    rng = np.random.RandomState(0)
    np.savez(
        path,
        obs=rng.randn(n, 3).astype(np.float32),
        action=np.tanh(rng.randn(n, 1)).astype(np.float32),
        reward=rng.randn(n).astype(np.float32),
        done=(rng.rand(n) < 0.02),
        next_obs=rng.randn(n, 3).astype(np.float32),
    )
    return path


@pytest.fixture(scope="module")
def cartpole_npz(tmp_path_factory):
    path = str(tmp_path_factory.mktemp("data") / "cartpole.npz")
    n = 256
    rng = np.random.RandomState(0)
    np.savez(
        path,
        obs=rng.randn(n, 4).astype(np.float32),
        action=rng.randint(0, 2, n),
        reward=rng.randn(n).astype(np.float32),
        done=(rng.rand(n) < 0.02),
        next_obs=rng.randn(n, 4).astype(np.float32),
    )
    return path


def _cfg(ptype, data_path, env='pendulum', model=None, extra=None):
    envs = {
        'pendulum': dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env'],
                         stop=-200, obs=3, act=1),
        'cartpole': dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env'],
                         stop=195, obs=4, act=2),
    }[env]
    main = EasyDict(dict(
        exp_name=f'exp/test_off_{ptype}',
        env=dict(evaluator_env_num=2, n_evaluator_episode=2, stop_value=envs['stop'], collector_env_num=1),
        policy=dict(
            cuda=False,
            model=model or dict(obs_shape=envs['obs'], action_shape=envs['act']),
            learn=dict(batch_size=32, update_per_collect=1),
            collect=dict(data_type='hdf5', data_path=data_path, unroll_len=1),
            eval=dict(evaluator=dict(eval_freq=int(1e6))),
        ),
    ))
    if extra:
        from ding.utils import deep_merge_dicts
        main.policy = EasyDict(deep_merge_dicts(main.policy, extra))
    create = EasyDict(dict(
        env=dict(type=envs['type'], import_names=envs['import_names']),
        env_manager=dict(type='base'),
        policy=dict(type=ptype),
    ))
    return main, create


def test_bc_continuous(pendulum_npz):
    main, create = _cfg('bc', pendulum_npz, extra=dict(continuous=True))
    serial_pipeline_offline((main, create), seed=0, max_train_iter=2)


def test_cql(pendulum_npz):
    main, create = _cfg('cql', pendulum_npz,
                        model=dict(obs_shape=3, action_shape=1, action_space='reparameterization', twin_critic=True))
    serial_pipeline_offline((main, create), seed=0, max_train_iter=2)


def test_td3_bc(pendulum_npz):
    main, create = _cfg('td3_bc', pendulum_npz,
                        model=dict(obs_shape=3, action_shape=1, action_space='regression', twin_critic=True))
    serial_pipeline_offline((main, create), seed=0, max_train_iter=2)


def test_iql(pendulum_npz):
    main, create = _cfg('iql', pendulum_npz,
                        model=dict(obs_shape=3, action_shape=1, action_space='reparameterization', twin_critic=True))
    serial_pipeline_offline((main, create), seed=0, max_train_iter=2)


def test_edac(pendulum_npz):
    main, create = _cfg('edac', pendulum_npz, model=dict(obs_shape=3, action_shape=1, ensemble_num=4))
    serial_pipeline_offline((main, create), seed=0, max_train_iter=2)


def test_discrete_cql(cartpole_npz):
    main, create = _cfg('discrete_cql', cartpole_npz, env='cartpole',
                        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[32, 32]),
                        extra=dict(nstep=1))
    serial_pipeline_offline((main, create), seed=0, max_train_iter=2)
