"""Env wrappers operating on gym-style envs (reset() -> obs,
step(a) -> (obs, reward, done, info)).

Parity: reference ding/envs/env_wrappers/env_wrappers.py:62-1490 (the 24
gym.Wrapper subclasses). Offline build notes: frame warping uses a numpy
area-resample (no cv2), and wrappers are registered in ENV_WRAPPER_REGISTRY
for cfg-driven composition through DingEnvWrapper.
"""
from collections import deque
from typing import Any, Optional, Union

import numpy as np

from ding.utils import ENV_WRAPPER_REGISTRY, RunningMeanStd
from ..common.spaces import Box, Discrete


class EnvWrapper:
    """Base delegating wrapper (gym.Wrapper analog, no gym dependency)."""

    def __init__(self, env):
        self.env = env

    def __getattr__(self, name):
        if name.startswith('_'):
            raise AttributeError(name)
        return getattr(self.env, name)

    def reset(self, **kwargs):
        return self.env.reset(**kwargs)

    def step(self, action):
        return self.env.step(action)

    @property
    def unwrapped(self):
        return getattr(self.env, 'unwrapped', self.env)


def _resize_area(frame: np.ndarray, size: tuple) -> np.ndarray:
    """Area-average resize for 2D (grayscale) arrays without cv2."""
    h, w = frame.shape[:2]
    th, tw = size
    ys = (np.linspace(0, h, th + 1)).astype(np.int64)
    xs = (np.linspace(0, w, tw + 1)).astype(np.int64)
    out = np.empty((th, tw), dtype=np.float32)
    f = frame.astype(np.float32)
    if f.ndim == 3:
        f = f.mean(-1)
    # integral image for O(1) block means
    integral = np.zeros((h + 1, w + 1), dtype=np.float64)
    integral[1:, 1:] = np.cumsum(np.cumsum(f, 0), 1)
    for i in range(th):
        y0, y1 = ys[i], max(ys[i + 1], ys[i] + 1)
        for j in range(tw):
            x0, x1 = xs[j], max(xs[j + 1], xs[j] + 1)
            s = integral[y1, x1] - integral[y0, x1] - integral[y1, x0] + integral[y0, x0]
            out[i, j] = s / ((y1 - y0) * (x1 - x0))
    return out


@ENV_WRAPPER_REGISTRY.register('noop_reset')
class NoopResetWrapper(EnvWrapper):
    """Random number of no-op steps at reset."""

    def __init__(self, env, noop_max: int = 30):
        super().__init__(env)
        self.noop_max = noop_max
        self.noop_action = 0

    def reset(self, **kwargs):
        obs = self.env.reset(**kwargs)
        noops = np.random.randint(1, self.noop_max + 1)
        for _ in range(noops):
            obs, reward, done, info = self.env.step(self.noop_action)
            if done:
                obs = self.env.reset(**kwargs)
        return obs


@ENV_WRAPPER_REGISTRY.register('max_and_skip')
class MaxAndSkipWrapper(EnvWrapper):
    """Repeat action ``skip`` times; obs = max over last two frames."""

    def __init__(self, env, skip: int = 4):
        super().__init__(env)
        self._skip = skip
        self._obs_buffer = deque(maxlen=2)

    def step(self, action):
        total_reward = 0.0
        done, info, obs = False, {}, None
        for _ in range(self._skip):
            obs, reward, done, info = self.env.step(action)
            self._obs_buffer.append(obs)
            total_reward += reward
            if done:
                break
        max_frame = np.max(np.stack(self._obs_buffer), axis=0)
        return max_frame, total_reward, done, info

    def reset(self, **kwargs):
        self._obs_buffer.clear()
        obs = self.env.reset(**kwargs)
        self._obs_buffer.append(obs)
        return obs


@ENV_WRAPPER_REGISTRY.register('warp_frame')
class WarpFrameWrapper(EnvWrapper):
    """Grayscale + resize to (size, size)."""

    def __init__(self, env, size: int = 84):
        super().__init__(env)
        self.size = size
        self.observation_space = Box(0, 255, (size, size), dtype=np.uint8)

    def _warp(self, frame):
        return _resize_area(frame, (self.size, self.size)).astype(np.uint8)

    def reset(self, **kwargs):
        return self._warp(self.env.reset(**kwargs))

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return self._warp(obs), reward, done, info


@ENV_WRAPPER_REGISTRY.register('scaled_float_frame')
class ScaledFloatFrameWrapper(EnvWrapper):

    def reset(self, **kwargs):
        return np.asarray(self.env.reset(**kwargs), dtype=np.float32) / 255.0

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return np.asarray(obs, dtype=np.float32) / 255.0, reward, done, info


@ENV_WRAPPER_REGISTRY.register('clip_reward')
class ClipRewardWrapper(EnvWrapper):

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return obs, float(np.sign(reward)), done, info


@ENV_WRAPPER_REGISTRY.register('frame_stack')
class FrameStackWrapper(EnvWrapper):
    """Stack last n frames along a new leading axis."""

    def __init__(self, env, n_frames: int = 4):
        super().__init__(env)
        self.n_frames = n_frames
        self.frames = deque(maxlen=n_frames)

    def reset(self, **kwargs):
        obs = self.env.reset(**kwargs)
        for _ in range(self.n_frames):
            self.frames.append(obs)
        return self._get_ob()

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        self.frames.append(obs)
        return self._get_ob(), reward, done, info

    def _get_ob(self):
        return np.stack(self.frames, axis=0)


@ENV_WRAPPER_REGISTRY.register('obs_norm')
class ObsNormWrapper(EnvWrapper):
    """Online observation normalization via RunningMeanStd."""

    def __init__(self, env):
        super().__init__(env)
        self.data_count = 0
        self.clip_range = (-3, 3)
        self.rms = RunningMeanStd(shape=())

    def _normalize(self, obs):
        obs = np.asarray(obs, dtype=np.float32)
        if self.data_count > 30:
            return np.clip((obs - self.rms.mean) / self.rms.std, *self.clip_range).astype(np.float32)
        return obs

    def reset(self, **kwargs):
        self.data_count = 0
        self.rms.reset()
        obs = np.asarray(self.env.reset(**kwargs), dtype=np.float32)
        self.data_count += 1
        self.rms.update(obs.reshape(1, -1).mean(-1, keepdims=True))
        return self._normalize(obs)

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        obs = np.asarray(obs, dtype=np.float32)
        self.data_count += 1
        self.rms.update(obs.reshape(1, -1).mean(-1, keepdims=True))
        return self._normalize(obs), reward, done, info


@ENV_WRAPPER_REGISTRY.register('reward_norm')
class RewardNormWrapper(EnvWrapper):
    """Normalize rewards by the std of the running discounted return."""

    def __init__(self, env, reward_discount: float = 0.99):
        super().__init__(env)
        self.cum_reward = np.zeros((1, ), dtype=np.float32)
        self.reward_discount = reward_discount
        self.rms = RunningMeanStd(shape=(1, ))
        self.data_count = 0

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        self.cum_reward = self.cum_reward * self.reward_discount + reward
        self.rms.update(self.cum_reward.reshape(1, 1))
        self.data_count += 1
        if self.data_count > 30:
            reward = float(reward / (self.rms.std[0] + 1e-8))
        return obs, reward, done, info

    def reset(self, **kwargs):
        self.cum_reward[:] = 0
        return self.env.reset(**kwargs)


@ENV_WRAPPER_REGISTRY.register('episodic_life')
class EpisodicLifeWrapper(EnvWrapper):
    """Treat life loss as episode end (info['lives'] protocol)."""

    def __init__(self, env):
        super().__init__(env)
        self.lives = 0
        self.was_real_done = True

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        self.was_real_done = done
        lives = info.get('lives', 0)
        if 0 < lives < self.lives:
            done = True
        self.lives = lives
        return obs, reward, done, info

    def reset(self, **kwargs):
        if self.was_real_done:
            obs = self.env.reset(**kwargs)
        else:
            obs, _, _, info = self.env.step(0)
        self.lives = 0
        return obs


@ENV_WRAPPER_REGISTRY.register('fire_reset')
class FireResetWrapper(EnvWrapper):
    """Press FIRE (action 1) after reset (Atari convention)."""

    def reset(self, **kwargs):
        self.env.reset(**kwargs)
        obs, _, done, _ = self.env.step(1)
        if done:
            obs = self.env.reset(**kwargs)
        return obs


@ENV_WRAPPER_REGISTRY.register('time_limit')
class TimeLimitWrapper(EnvWrapper):

    def __init__(self, env, max_limit: int = 1000):
        super().__init__(env)
        self.max_limit = max_limit
        self._elapsed = 0

    def reset(self, **kwargs):
        self._elapsed = 0
        return self.env.reset(**kwargs)

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        self._elapsed += 1
        if self._elapsed >= self.max_limit:
            done = True
            info['time_limit'] = True
        return obs, reward, done, info


@ENV_WRAPPER_REGISTRY.register('delay_reward')
class DelayRewardWrapper(EnvWrapper):
    """Accumulate rewards and release every ``delay_reward_step`` steps."""

    def __init__(self, env, delay_reward_step: int = 0):
        super().__init__(env)
        self._delay_step = delay_reward_step
        self._delay_buffer = 0.0
        self._delay_count = 0

    def reset(self, **kwargs):
        self._delay_buffer, self._delay_count = 0.0, 0
        return self.env.reset(**kwargs)

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        if self._delay_step <= 1:
            return obs, reward, done, info
        self._delay_buffer += reward
        self._delay_count += 1
        if self._delay_count >= self._delay_step or done:
            reward, self._delay_buffer, self._delay_count = self._delay_buffer, 0.0, 0
        else:
            reward = 0.0
        return obs, reward, done, info


@ENV_WRAPPER_REGISTRY.register('eval_episode_return')
class EvalEpisodeReturnWrapper(EnvWrapper):
    """Accumulate raw return; writes info['eval_episode_return'] at done."""

    def reset(self, **kwargs):
        self._eval_episode_return = 0.0
        return self.env.reset(**kwargs)

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        self._eval_episode_return += float(reward)
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return obs, reward, done, info


@ENV_WRAPPER_REGISTRY.register('obs_transpose')
class ObsTransposeWrapper(EnvWrapper):
    """HWC -> CHW."""

    def _t(self, obs):
        return np.transpose(obs, (2, 0, 1))

    def reset(self, **kwargs):
        return self._t(self.env.reset(**kwargs))

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return self._t(obs), reward, done, info


@ENV_WRAPPER_REGISTRY.register('flat_obs')
class FlatObsWrapper(EnvWrapper):

    def reset(self, **kwargs):
        return np.asarray(self.env.reset(**kwargs)).reshape(-1)

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return np.asarray(obs).reshape(-1), reward, done, info


@ENV_WRAPPER_REGISTRY.register('action_repeat')
class ActionRepeatWrapper(EnvWrapper):

    def __init__(self, env, action_repeat: int = 1):
        super().__init__(env)
        self.action_repeat = action_repeat

    def step(self, action):
        reward = 0.0
        for _ in range(self.action_repeat):
            obs, r, done, info = self.env.step(action)
            reward += r
            if done:
                break
        return obs, reward, done, info


def update_shape(obs_shape, act_shape, rew_shape, wrapper_names):
    """Best-effort static shape propagation through a wrapper list."""
    for name in wrapper_names:
        if name == 'warp_frame':
            obs_shape = (84, 84)
        elif name == 'frame_stack':
            obs_shape = (4, *obs_shape)
    return obs_shape, act_shape, rew_shape


@ENV_WRAPPER_REGISTRY.register('static_obs_norm')
class StaticObsNormWrapper(EnvWrapper):
    """Normalize observations with a PRECOMPUTED dataset mean/std
    (reference env_wrappers.py StaticObsNormWrapper:790)."""

    def __init__(self, env, mean: np.ndarray, std: np.ndarray):
        super().__init__(env)
        self.mean = np.asarray(mean, dtype=np.float32)
        self.std = np.asarray(std, dtype=np.float32)
        self.clip_range = (-3, 3)

    def _norm(self, obs):
        return np.clip((np.asarray(obs, dtype=np.float32) - self.mean) / (self.std + 1e-8), *self.clip_range)

    def reset(self, **kwargs):
        return self._norm(self.env.reset(**kwargs))

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return self._norm(obs), reward, done, info


@ENV_WRAPPER_REGISTRY.register('ram')
class RamWrapper(EnvWrapper):
    """Reshape a RAM vector observation into an image-like [N, 1, 1] tensor
    (reference RamWrapper:912)."""

    def reset(self, **kwargs):
        obs = self.env.reset(**kwargs)
        return np.asarray(obs, dtype=np.float32).reshape(-1, 1, 1)

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return np.asarray(obs, dtype=np.float32).reshape(-1, 1, 1), reward, done, info


@ENV_WRAPPER_REGISTRY.register('obs_plus_prev_action_reward')
@ENV_WRAPPER_REGISTRY.register('obs_plus_prev_act_rew')
class ObsPlusPrevActRewWrapper(EnvWrapper):
    """NGU input contract: obs dict {obs, prev_action, prev_reward_extrinsic}
    (reference ObsPlusPrevActRewWrapper:1144)."""

    def __init__(self, env):
        super().__init__(env)
        self.prev_action = -1
        self.prev_reward_extrinsic = 0.0

    def reset(self, **kwargs):
        obs = self.env.reset(**kwargs)
        self.prev_action = -1
        self.prev_reward_extrinsic = 0.0
        return {'obs': obs, 'prev_action': self.prev_action, 'prev_reward_extrinsic': self.prev_reward_extrinsic}

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        out = {'obs': obs, 'prev_action': self.prev_action, 'prev_reward_extrinsic': self.prev_reward_extrinsic}
        self.prev_action = action
        self.prev_reward_extrinsic = float(np.asarray(reward).reshape(-1)[0])
        return out, reward, done, info


@ENV_WRAPPER_REGISTRY.register('transpose')
class TransposeWrapper(EnvWrapper):
    """HWC -> CHW observation transpose (reference TransposeWrapper:1213)."""

    def _process_obs(self, obs):
        return np.transpose(np.asarray(obs), (2, 0, 1))

    def reset(self, **kwargs):
        return self._process_obs(self.env.reset(**kwargs))

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return self._process_obs(obs), reward, done, info


@ENV_WRAPPER_REGISTRY.register('gym_to_gymnasium')
class GymToGymnasiumWrapper(EnvWrapper):
    """Adapt a gymnasium-API env (reset->(obs, info), step->5-tuple) to the
    gym 4-tuple API (reference GymToGymnasiumWrapper:1436)."""

    def __init__(self, env):
        super().__init__(env)
        self._seed = None

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed

    def reset(self, **kwargs):
        if self._seed is not None:
            out = self.env.reset(seed=self._seed, **kwargs)
        else:
            out = self.env.reset(**kwargs)
        return out[0] if isinstance(out, tuple) else out

    def step(self, action):
        out = self.env.step(action)
        if len(out) == 5:
            obs, reward, terminated, truncated, info = out
            return obs, reward, terminated or truncated, info
        return out


@ENV_WRAPPER_REGISTRY.register('reward_in_obs')
@ENV_WRAPPER_REGISTRY.register('all_in_obs')
class AllinObsWrapper(EnvWrapper):
    """Decision-Transformer input contract: obs dict {obs, reward}
    (reference AllinObsWrapper:1490)."""

    def reset(self, **kwargs):
        obs = self.env.reset(**kwargs)
        return {'obs': obs, 'reward': np.zeros(1, dtype=np.float32)}

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        return {'obs': obs, 'reward': np.asarray(reward, dtype=np.float32).reshape(-1)}, reward, done, info


@ENV_WRAPPER_REGISTRY.register('gym_hybrid_dict_action')
class GymHybridDictActionWrapper(EnvWrapper):
    """Tuple -> Dict action adaptation for gym-hybrid envs
    (reference GymHybridDictActionWrapper:1077)."""

    def step(self, action):
        if isinstance(action, dict):
            action = (action.get('type', action.get('action_type')), action.get('args', action.get('action_args')))
        return self.env.step(action)
