"""POMDP Atari (RAM observations with observation corruption).

Parity with the reference dizoo/pomdp/envs/atari_env.py: RAM-style flat
observations (512 = 4-stack of 128 RAM bytes) corrupted by the reference's
four POMDP knobs — additive ``noise_scale`` gaussian noise, ``zero_p``
full-observation dropout (flicker), ``reward_noise``, and ``duplicate_p``
(repeat the previous observation instead of the fresh one). The underlying
game is the atari-lite moving-target task encoded into RAM bytes, so
recurrent policies (R2D2/DRQN) can exploit memory where feedforward ones
plateau.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('pomdp')
class PomdpLiteEnv(BaseEnv):

    RAM = 128

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        self._cfg = cfg
        self.frame_stack = cfg.get('frame_stack', 4)
        self.action_num = cfg.get('action_num', 6)
        self.max_step = cfg.get('max_step', 400)
        pomdp = cfg.get('pomdp', {})
        self.noise_scale = pomdp.get('noise_scale', 0.01)
        self.zero_p = pomdp.get('zero_p', 0.2)
        self.reward_noise = pomdp.get('reward_noise', 0.01)
        self.duplicate_p = pomdp.get('duplicate_p', 0.2)
        self.obs_dim = self.RAM * self.frame_stack
        self._observation_space = Box(0.0, 1.0, (self.obs_dim, ))
        self._action_space = Discrete(self.action_num)
        self._reward_space = Box(-1, 1, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def _ram(self) -> np.ndarray:
        """Encode the hidden game state into 128 'RAM bytes' (normalized)."""
        ram = (self._rng.rand(self.RAM) * 0.1).astype(np.float32)
        ram[0] = self._pos[0] / 84.0
        ram[1] = self._pos[1] / 84.0
        ram[2] = (self._vel[0] + 4) / 8.0
        ram[3] = (self._vel[1] + 4) / 8.0
        return ram

    def _corrupt(self, ram: np.ndarray) -> np.ndarray:
        if self._rng.rand() < self.duplicate_p and self._last_ram is not None:
            ram = self._last_ram.copy()
        if self._rng.rand() < self.zero_p:
            ram = np.zeros_like(ram)
        ram = ram + self._rng.randn(self.RAM).astype(np.float32) * self.noise_scale
        return ram.astype(np.float32)

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._pos = self._rng.rand(2) * 84
        self._vel = self._rng.randn(2) * 2
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._last_ram = None
        first = self._ram()
        self._last_ram = first
        self._frames = [self._corrupt(first) for _ in range(self.frame_stack)]
        return np.concatenate(self._frames)

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.reshape(-1)[0])
        action = int(action)
        quadrant = (int(self._pos[0] > 42) * 2 + int(self._pos[1] > 42)) % self.action_num
        reward = 1.0 if action == quadrant else -0.05
        reward += float(self._rng.randn()) * self.reward_noise
        self._pos = (self._pos + self._vel) % 84
        if self._rng.rand() < 0.05:
            self._vel = self._rng.randn(2) * 2
        ram = self._ram()
        self._frames.pop(0)
        self._frames.append(self._corrupt(ram))
        self._last_ram = ram
        self._step_count += 1
        self._eval_episode_return += reward
        done = self._step_count >= self.max_step
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(
            np.concatenate(self._frames), np.array([reward], dtype=np.float32), done, info
        )

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return np.array([self._action_space.sample()], dtype=np.int64)

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
        return "PomdpLiteEnv"
