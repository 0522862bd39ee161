"""Super-Mario-style side-scroller ("mario-lite", reference dizoo/mario
wrapping gym-super-mario-bros). Scrolling terrain with gaps and blocks
rendered to the 4x84x84 frame-stack interface; actions {noop, right,
right+jump}; reward = forward progress, death on falling into a gap;
flag at x = level_len ends the level with +15 (the reference's reward
scale).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('mario')
class MarioLiteEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.level_len = int(self._cfg.get('level_len', 200))
        self._max_step = self._cfg.get('max_step', 500)
        self.frame_stack = 4
        self._observation_space = Box(0.0, 1.0, (4, 84, 84))
        self._action_space = Discrete(3)
        self._reward_space = Box(-15.0, 15.0, (1, ))
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
        # terrain: ground height per column, 0 marks a gap
        self.ground = np.full(self.level_len + 20, 20.0)
        x = 10
        while x < self.level_len - 10:
            x += self._rng.randint(8, 20)
            gap = self._rng.randint(2, 4)
            self.ground[x:x + gap] = 0.0
            x += gap
        self.x = 2.0
        self.y = 20.0
        self.vy = 0.0
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._frames = [self._frame() for _ in range(self.frame_stack)]
        return np.stack(self._frames)

    def _frame(self) -> np.ndarray:
        f = np.zeros((84, 84), dtype=np.float32)
        x0 = int(self.x) - 10
        for col in range(84 // 4):
            gx = x0 + col
            if 0 <= gx < len(self.ground) and self.ground[gx] > 0:
                h = int(self.ground[gx])
                f[84 - h:, col * 4:(col + 1) * 4] = 0.5
        ay = int(np.clip(84 - self.y - 4, 0, 80))
        f[ay:ay + 4, 40:44] = 1.0
        return f

    def step(self, action: Any) -> BaseEnvTimestep:
        if hasattr(action, 'reshape'):
            action = int(np.asarray(action).reshape(-1)[0])
        action = int(action)
        dt = 0.5
        gx = int(self.x)
        on_ground = self.ground[gx] > 0 and self.y <= self.ground[gx] + 0.1
        vx = 1.0 if action in (1, 2) else 0.0
        if action == 2 and on_ground:
            self.vy = 4.0
        self.vy -= 1.0 * dt
        prev_x = self.x
        self.x += vx * dt
        self.y += self.vy * dt
        gx = int(self.x)
        if self.ground[gx] > 0 and self.y <= self.ground[gx]:
            self.y = self.ground[gx]
            self.vy = 0.0
        reward = (self.x - prev_x)
        done = False
        if self.y < -2.0:  # fell into a gap
            reward = -15.0
            done = True
        if self.x >= self.level_len:  # flag
            reward += 15.0
            done = True
        self._step_count += 1
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        self._frames.pop(0)
        self._frames.append(self._frame())
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(np.stack(self._frames), np.array([reward], dtype=np.float32), done, info)

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
        return "MarioLiteEnv"
