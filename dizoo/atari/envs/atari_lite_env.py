"""Synthetic Atari-shaped env ("atari-lite"): 4x84x84 uint8 frame stack,
discrete actions, reward structure with a learnable signal.

The real ALE binaries are not available offline; this env preserves the
observation/action/reward interface and compute shape of the reference's
Atari pipeline (dizoo/atari/envs/atari_env.py) so Pong/SpaceInvaders configs
run end to end and exercise the conv path. The hidden dynamics are a
moving-target game: a bright patch drifts across the frame and the correct
action depends on its quadrant, so policies can actually improve.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('atari_lite')
class AtariLiteEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = cfg or {}
        self._cfg = cfg
        self.frame_stack = cfg.get('frame_stack', 4)
        self.size = cfg.get('size', 84)
        self.action_num = cfg.get('action_num', 6)
        self.max_step = cfg.get('max_step', 400)
        self._observation_space = Box(0, 255, (self.frame_stack, self.size, self.size), dtype=np.uint8)
        self._action_space = Discrete(self.action_num)
        self._reward_space = Box(-1, 1, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def _frame(self) -> np.ndarray:
        f = (self._rng.rand(self.size, self.size) * 30).astype(np.uint8)
        y, x = int(self._pos[0]), int(self._pos[1])
        f[max(0, y - 3):y + 3, max(0, x - 3):x + 3] = 255
        return f

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._pos = self._rng.rand(2) * self.size
        self._vel = self._rng.randn(2) * 2
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._frames = [self._frame() for _ in range(self.frame_stack)]
        return np.stack(self._frames)

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.reshape(-1)[0])
        action = int(action)
        # correct action = quadrant of the target (mod action_num)
        quadrant = (int(self._pos[0] > self.size / 2) * 2 + int(self._pos[1] > self.size / 2)) % self.action_num
        reward = 1.0 if action == quadrant else -0.05
        self._pos = (self._pos + self._vel) % self.size
        if self._rng.rand() < 0.05:
            self._vel = self._rng.randn(2) * 2
        self._frames.pop(0)
        self._frames.append(self._frame())
        self._step_count += 1
        self._eval_episode_return += reward
        done = self._step_count >= self.max_step
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(
            np.stack(self._frames), np.array([reward], dtype=np.float32), done, info
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
        return "AtariLiteEnv"
