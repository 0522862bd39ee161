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


def check_space_dtype(env: BaseEnv) -> None:
    """Float spaces must be fp32, int spaces int64 (reference checker step 0)."""
    env.reset()
    for name, space in zip(['obs', 'act', 'rew'], [env.observation_space, env.action_space, env.reward_space]):
        dt = getattr(space, 'dtype', None)
        if dt is None:
            continue
        if 'float' in repr(dt):
            assert np.dtype(dt) == np.float32, f"float {name} space must be np.float32, got {dt}"
        elif 'int' in repr(dt):
            assert np.dtype(dt) == np.int64, f"int {name} space must be np.int64, got {dt}"


def check_array_space(data, space, name: str) -> None:
    """Assert data matches the (possibly nested) space's dtype/shape/bounds."""
    if isinstance(data, np.ndarray):
        if getattr(space, 'dtype', None) is not None:
            assert data.dtype == space.dtype, f"{name} dtype {data.dtype} != space {space.dtype}"
        if getattr(space, 'shape', None):
            assert data.shape == tuple(space.shape), f"{name} shape {data.shape} != space {space.shape}"
        low, high = getattr(space, 'low', None), getattr(space, 'high', None)
        if low is not None and high is not None:
            assert (low <= data).all() and (data <= high).all(), f"{name} out of Box bounds"
        n = getattr(space, 'n', None)
        if n is not None:
            assert (np.asarray(data) >= 0).all() and (np.asarray(data) < n).all(), f"{name} out of Discrete range"
    elif isinstance(data, dict):
        sub = getattr(space, 'spaces', space)
        for k, v in data.items():
            if isinstance(sub, dict) and k in sub:
                check_array_space(v, sub[k], f'{name}.{k}')
    elif isinstance(data, (list, tuple)):
        for i, v in enumerate(data):
            s = space[i] if isinstance(space, (list, tuple)) else space
            check_array_space(v, s, f'{name}[{i}]')


def check_reset(env: BaseEnv) -> None:
    obs = env.reset()
    check_array_space(obs, env.observation_space, 'obs')


def check_step(env: BaseEnv, max_done: int = 3, step_cap: int = 10000) -> None:
    """Step with random actions through ``max_done`` episode ends, checking
    obs/rew against their spaces and the eval_episode_return contract."""
    env.reset()
    done_times = 0
    for _ in range(step_cap):
        action = env.random_action() if hasattr(env, 'random_action') else env.action_space.sample()
        ts = env.step(action)
        obs, rew, done, info = ts.obs, ts.reward, ts.done, ts.info
        check_array_space(obs, env.observation_space, 'obs')
        if isinstance(rew, np.ndarray):
            check_array_space(rew, env.reward_space, 'rew')
        if done:
            assert 'eval_episode_return' in info, "done timestep must carry info['eval_episode_return']"
            done_times += 1
            env.reset()
        if done_times >= max_done:
            return
    raise AssertionError(f"no {max_done} episode ends within {step_cap} steps")


def check_different_memory(array1, array2, step_times: int) -> None:
    """Consecutive observations must not alias the same buffers."""
    assert type(array1) == type(array2), f"step {step_times}: obs types differ"
    if isinstance(array1, np.ndarray):
        assert id(array1) != id(array2), f"step {step_times}: consecutive obs share one ndarray (missing copy)"
    elif isinstance(array1, dict):
        assert array1.keys() == array2.keys()
        for k in array1:
            check_different_memory(array1[k], array2[k], step_times)
    elif isinstance(array1, (list, tuple)):
        assert len(array1) == len(array2)
        for a, b in zip(array1, array2):
            check_different_memory(a, b, step_times)
    elif not isinstance(array1, (int, float, np.number)):
        raise TypeError(f"unsupported obs type: {type(array1)}")


def check_obs_deepcopy(env: BaseEnv, step_cap: int = 10000) -> None:
    obs_prev = env.reset()
    for t in range(1, step_cap + 1):
        action = env.random_action() if hasattr(env, 'random_action') else env.action_space.sample()
        ts = env.step(action)
        check_different_memory(obs_prev, ts.obs, t)
        obs_prev = ts.obs
        if ts.done:
            return


def check_all(env: BaseEnv) -> None:
    check_space_dtype(env)
    check_reset(env)
    check_step(env)
    check_obs_deepcopy(env)


def demonstrate_correct_procedure(env_fn) -> None:
    """Reference usage pattern: seed before reset, reset re-arms on done."""
    env = env_fn({})
    env.seed(4)
    env.reset()
    done_times = 0
    while done_times < 2:
        ts = env.step(env.random_action())
        if ts.done:
            assert 'eval_episode_return' in ts.info
            done_times += 1
            env.reset()
    env.close()
