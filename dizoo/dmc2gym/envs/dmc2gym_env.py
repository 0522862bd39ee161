"""dm_control suite tasks via the dmc2gym interface, implemented natively
(dm_control unavailable offline). Parity with the reference
dizoo/dmc2gym/envs/dmc2gym_env.py surface: cfg(domain_name, task_name,
from_pixels, frame_skip), continuous Box actions in [-1, 1], per-step reward
in [0, 1], 1000-step episodes (returns up to 1000).

* cartpole balance / swingup: real cart-pole continuous-force dynamics with
  the dm_control-style smooth reward (upright cosine x centering), state obs
  [cos th, sin th, x, x_dot, th_dot] or 3x84x84 rendered pixels.
* cheetah run / walker walk: mujoco-lite-style smooth latent dynamics at the
  dm_control shapes (17/6 and 24/6), reward = squashed forward progress.
"""
import math
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('dmc2gym')
class DMC2GymEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.domain = self._cfg.get('domain_name', 'cartpole')
        self.task = self._cfg.get('task_name', 'balance')
        self.from_pixels = self._cfg.get('from_pixels', False)
        self.frame_skip = int(self._cfg.get('frame_skip', 1))
        self.channels_first = self._cfg.get('channels_first', True)
        self._max_step = self._cfg.get('max_step', 1000 // max(self.frame_skip, 1))
        if self.domain == 'cartpole':
            self.obs_dim, self.act_dim = 5, 1
        elif self.domain == 'cheetah':
            self.obs_dim, self.act_dim = 17, 6
        else:  # walker and friends
            self.obs_dim, self.act_dim = 24, 6
        if self.from_pixels:
            self._observation_space = Box(0.0, 1.0, (3, 84, 84))
        else:
            self._observation_space = Box(-np.inf, np.inf, (self.obs_dim, ))
        self._action_space = Box(-1.0, 1.0, (self.act_dim, ))
        self._reward_space = Box(0.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        if self.domain == 'cartpole':
            if self.task == 'swingup':
                theta = math.pi + self._rng.uniform(-0.1, 0.1)  # hanging down
            else:
                theta = self._rng.uniform(-0.05, 0.05)
            self._state = np.array([0.0, 0.0, theta, 0.0])  # x, x_dot, th, th_dot
        else:
            rng = np.random.RandomState(54321)
            self._A = rng.randn(self.obs_dim, self.obs_dim) * 0.1
            self._A /= max(1.0, np.abs(np.linalg.eigvals(self._A)).max() / 0.95)
            self._B = rng.randn(self.obs_dim, self.act_dim) * 0.5
            self._w = rng.randn(self.obs_dim) / np.sqrt(self.obs_dim)
            self._state = self._rng.randn(self.obs_dim) * 0.1
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        if self.domain == 'cartpole':
            x, x_dot, th, th_dot = self._state
            state = np.array([math.cos(th), math.sin(th), x, x_dot, th_dot], dtype=np.float32)
        else:
            state = self._state.astype(np.float32)
        if not self.from_pixels:
            return state
        return self._render()

    def _render(self) -> np.ndarray:
        """84x84 cart-pole render (pixel lane only supports cartpole)."""
        img = np.zeros((3, 84, 84), dtype=np.float32)
        x, _, th, _ = self._state if self.domain == 'cartpole' else (0, 0, 0, 0)
        cx = int(np.clip(42 + x * 15, 6, 77))
        img[2, 60:64, max(cx - 6, 0):cx + 6] = 1.0  # cart blue
        tip_r = 60 - int(28 * math.cos(th))
        tip_c = cx + int(28 * math.sin(th))
        rr = np.linspace(60, tip_r, 28).astype(int).clip(0, 83)
        cc = np.linspace(cx, tip_c, 28).astype(int).clip(0, 83)
        img[0, rr, cc] = 1.0  # pole red
        return img

    def _cartpole_step(self, force: float) -> None:
        g, mc, mp, l, dt = 9.8, 1.0, 0.1, 0.5, 0.01
        x, x_dot, th, th_dot = self._state
        f = 10.0 * force
        cos, sin = math.cos(th), math.sin(th)
        tmp = (f + mp * l * th_dot ** 2 * sin) / (mc + mp)
        th_acc = (g * sin - cos * tmp) / (l * (4.0 / 3.0 - mp * cos ** 2 / (mc + mp)))
        x_acc = tmp - mp * l * th_acc * cos / (mc + mp)
        x += dt * x_dot
        x_dot += dt * x_acc
        x = float(np.clip(x, -2.4, 2.4))
        th += dt * th_dot
        th_dot += dt * th_acc
        self._state = np.array([x, x_dot, th, th_dot])

    def step(self, action: Any) -> BaseEnvTimestep:
        a = np.clip(np.asarray(action, dtype=np.float64).reshape(-1), -1, 1)
        reward = 0.0
        for _ in range(self.frame_skip):
            if self.domain == 'cartpole':
                self._cartpole_step(float(a[0]))
                x, _, th, th_dot = self._state
                upright = (math.cos(th) + 1.0) / 2.0        # 1 when upright
                centered = 1.0 - min(abs(x) / 2.4, 1.0)
                small_vel = 1.0 / (1.0 + 0.1 * th_dot ** 2)
                reward += upright * (0.8 + 0.2 * centered) * small_vel
            else:
                self._state = self._A @ self._state + self._B @ a + self._rng.randn(self.obs_dim) * 0.01
                self._state = np.clip(self._state, -10, 10)
                progress = float(self._w @ self._state)
                reward += 1.0 / (1.0 + math.exp(-progress))  # squashed to (0, 1)
        reward /= self.frame_skip
        self._step_count += 1
        self._eval_episode_return += reward
        done = self._step_count >= self._max_step
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return self._action_space.sample().astype(np.float32)

    @property
    def observation_space(self):
        return self._observation_space

    @property
    def action_space(self):
        return self._action_space

    @property
    def reward_space(self):
        return self._reward_space

    def __repr__(self) -> str:
        return f"DMC2GymEnv({self.domain}-{self.task})"
