"""User-env conformance checker.

Parity: reference ding/envs/env/env_implementation_check.py:192.
"""
import numpy as np

from .base_env import BaseEnv, BaseEnvTimestep


def check_env_implementation(env: BaseEnv, n_steps: int = 10) -> bool:
    """Run a short reset/step/close cycle asserting the BaseEnv contract."""
    obs = env.reset()
    assert obs is not None, "reset() must return the initial observation"
    space = getattr(env, 'observation_space', None)
    if space is not None and getattr(space, 'shape', None):
        assert tuple(np.asarray(obs).shape) == tuple(space.shape), \
            f"obs shape {np.asarray(obs).shape} != observation_space {space.shape}"
    for _ in range(n_steps):
        action = env.random_action()
        ts = env.step(action)
        assert isinstance(ts, BaseEnvTimestep), "step() must return a BaseEnvTimestep"
        assert isinstance(ts.info, dict)
        if ts.done:
            assert 'eval_episode_return' in ts.info, "done timestep must report info['eval_episode_return']"
            env.reset()
    env.close()
    return True
