"""Behavioral failure-injection matrix for the subprocess env managers
(reference pattern: ding/envs/env_manager/tests/ async-reset / watchdog /
retry-renew / shm coverage).
"""
import numpy as np
import pytest

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import EasyDict


class FaultyEnv(BaseEnv):
    """Configurable failure env: crashes or hangs on chosen reset/step calls.

    cfg:
      crash_on_reset_attempts: int — raise on the first N reset() calls
      hang_on_step: int — step index that sleeps forever (watchdog food)
      crash_on_step: int — step index that raises
    """

    def __init__(self, cfg=None):
        cfg = cfg or {}
        self._crash_resets = cfg.get('crash_on_reset_attempts', 0)
        self._hang_step = cfg.get('hang_on_step', -1)
        self._crash_step = cfg.get('crash_on_step', -1)
        self._episode_len = cfg.get('episode_len', 5)
        self._reset_count = 0
        self._step_count = 0
        self._seed = 0
        self._observation_space = Box(-1, 1, (4, ))
        self._action_space = Discrete(2)
        self._reward_space = Box(-1, 1, (1, ))

    def reset(self):
        self._reset_count += 1
        if self._reset_count <= self._crash_resets:
            raise RuntimeError(f"injected reset crash #{self._reset_count}")
        self._step_count = 0
        return np.full(4, float(self._seed % 7), dtype=np.float32)

    def step(self, action):
        self._step_count += 1
        if self._step_count == self._hang_step:
            import time
            time.sleep(3600)
        if self._step_count == self._crash_step:
            raise RuntimeError("injected step crash")
        done = self._step_count >= self._episode_len
        info = {'eval_episode_return': float(self._step_count)} if done else {}
        return BaseEnvTimestep(
            np.full(4, float(self._step_count), dtype=np.float32),
            np.array([1.0], dtype=np.float32), done, info
        )

    def seed(self, seed, dynamic_seed=True):
        self._seed = seed

    def close(self):
        pass

    def random_action(self):
        return np.array([0], dtype=np.int64)

    @property
    def observation_space(self):
        return self._observation_space

    @property
    def action_space(self):
        return self._action_space

    @property
    def reward_space(self):
        return self._reward_space

    def __repr__(self):
        return "FaultyEnv"


def _make(cfg_env=None, n=2, **mgr_kwargs):
    from ding.envs.env_manager.subprocess_env_manager import SyncSubprocessEnvManager
    cfg = EasyDict({**SyncSubprocessEnvManager.default_config(), **mgr_kwargs})
    return SyncSubprocessEnvManager([lambda c=cfg_env: FaultyEnv(c) for _ in range(n)], cfg)


def test_reset_retry_renew_recovers():
    """A reset that crashes once per env is absorbed by retry_type='reset'
    (same worker retried; 'renew' would hand the fresh worker the same
    first-reset crash)."""
    mgr = _make({'crash_on_reset_attempts': 1}, max_retry=3, retry_type='reset', shared_memory=False)
    try:
        mgr.launch()
        obs = mgr.ready_obs
        assert len(obs) == 2
    finally:
        mgr.close()


def test_reset_gives_up_after_max_retry():
    mgr = _make({'crash_on_reset_attempts': 100}, max_retry=2, shared_memory=False, reset_timeout=5)
    with pytest.raises(RuntimeError):
        mgr.launch()
    mgr.close()


def test_step_watchdog_timeout():
    """A hung step trips the step_timeout watchdog instead of blocking."""
    mgr = _make({'hang_on_step': 2}, step_timeout=1.0, shared_memory=False)
    try:
        mgr.launch()
        mgr.step(mgr.random_action())  # step 1 fine
        with pytest.raises((TimeoutError, RuntimeError)):
            mgr.step(mgr.random_action())  # step 2 hangs -> watchdog
    finally:
        mgr.close()


def test_step_crash_surfaces():
    mgr = _make({'crash_on_step': 1}, shared_memory=False, step_timeout=5)
    try:
        mgr.launch()
        with pytest.raises(RuntimeError):
            mgr.step(mgr.random_action())
    finally:
        mgr.close()


def test_shared_memory_obs_roundtrip():
    """shm lane returns the same obs values as the pipe lane."""
    mgr_shm = _make(None, shared_memory=True)
    mgr_pipe = _make(None, shared_memory=False)
    try:
        mgr_shm.seed([7, 7])
        mgr_pipe.seed([7, 7])
        mgr_shm.launch()
        mgr_pipe.launch()
        o1 = {i: v for i, v in mgr_shm.ready_obs.items()}
        o2 = {i: v for i, v in mgr_pipe.ready_obs.items()}
        for i in o1:
            assert np.allclose(o1[i], o2[i])
        t1 = mgr_shm.step({i: np.array([0]) for i in o1})
        t2 = mgr_pipe.step({i: np.array([0]) for i in o2})
        for i in t1:
            assert np.allclose(t1[i].obs, t2[i].obs)
            assert t1[i].reward == t2[i].reward
    finally:
        mgr_shm.close()
        mgr_pipe.close()


def test_auto_reset_lifecycle():
    """done -> child auto-resets; ready_obs carries the fresh reset obs."""
    mgr = _make({'episode_len': 2}, shared_memory=False)
    try:
        mgr.launch()
        acts = mgr.random_action()
        mgr.step(acts)
        ts = mgr.step(mgr.random_action())  # episode ends here
        for i, t in ts.items():
            assert t.done
            assert 'eval_episode_return' in t.info
        # next obs must be the reset obs (step counter back to 0 encoding)
        for i, o in mgr.ready_obs.items():
            assert np.allclose(o, 0.0), o
        mgr.step(mgr.random_action())  # stepping after auto-reset works
    finally:
        mgr.close()


def test_async_manager_partial_ready():
    from ding.envs.env_manager.subprocess_env_manager import AsyncSubprocessEnvManager
    cfg = EasyDict({**AsyncSubprocessEnvManager.default_config(),
                    'shared_memory': False, 'wait_num': 1})
    mgr = AsyncSubprocessEnvManager([lambda: FaultyEnv({'episode_len': 50}) for _ in range(3)], cfg)
    try:
        mgr.launch()
        ready = mgr.ready_obs
        assert 1 <= len(ready) <= 3
        mgr.step({i: np.array([0]) for i in ready})
        ready2 = mgr.ready_obs
        assert len(ready2) >= 1  # wait_num=1: returns as soon as one env lands
    finally:
        mgr.close()


def test_seed_determinism_across_subprocess():
    a, b = _make(None, shared_memory=False), _make(None, shared_memory=False)
    try:
        a.seed([3, 4])
        b.seed([3, 4])
        a.launch()
        b.launch()
        oa, ob = a.ready_obs, b.ready_obs
        for i in oa:
            assert np.allclose(oa[i], ob[i])
    finally:
        a.close()
        b.close()
