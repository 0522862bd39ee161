"""Competitive 1v1 Pong (reference dizoo/competitive_rl wrapping
competitive_rl's cPong). State-based offline implementation: ball + two
paddles; obs 8 per side (own paddle y/vy, ball x/y/vx/vy, opponent y/vy),
x mirrored per side; Discrete(3) stay/up/down; +-1 per point, first to 5.
``opponent='builtin'`` plays a tracking bot; ``opponent='agent'`` takes a
list of two actions and returns per-side obs/reward (battle lane).
"""
from typing import Any, List, Union

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

DT = 1.0 / 30.0
PADDLE_SPEED = 1.2
BALL_SPEED = 1.0


@ENV_REGISTRY.register('competitive_pong')
class CompetitivePongEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._vs_agent = self._cfg.get('opponent', 'builtin') == 'agent'
        self._max_step = self._cfg.get('max_step', 1500)
        self._points = self._cfg.get('points', 5)
        self._observation_space = Box(-np.inf, np.inf, (8, ))
        self._action_space = Discrete(3)
        self._reward_space = Box(-1.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def _serve(self, direction: int) -> None:
        ang = self._rng.uniform(-0.6, 0.6)
        self.ball = np.array([0.0, self._rng.uniform(-0.3, 0.3),
                              BALL_SPEED * direction * np.cos(ang), BALL_SPEED * np.sin(ang)])

    def reset(self) -> Union[np.ndarray, List[np.ndarray]]:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self.p = np.zeros(2)   # paddle y, left(0) right(1)
        self.pv = np.zeros(2)
        self.score = [0, 0]
        self._serve(1 if self._rng.rand() < 0.5 else -1)
        self._step_count = 0
        self._eval_episode_return = 0.0
        if self._vs_agent:
            return [self._obs(0), self._obs(1)]
        return self._obs(0)

    def _obs(self, side: int) -> np.ndarray:
        m = 1.0 if side == 0 else -1.0  # mirror x for side 1
        me, op = (0, 1) if side == 0 else (1, 0)
        return np.array([
            self.p[me], self.pv[me], m * self.ball[0], self.ball[1], m * self.ball[2], self.ball[3],
            self.p[op], self.pv[op]
        ], dtype=np.float32)

    def _move(self, side: int, action: int) -> None:
        v = {0: 0.0, 1: PADDLE_SPEED, 2: -PADDLE_SPEED}[int(action)]
        self.pv[side] = v
        self.p[side] = float(np.clip(self.p[side] + v * DT, -1.0, 1.0))

    def _bot(self) -> int:
        target = self.ball[1]
        if target > self.p[1] + 0.05:
            return 1
        if target < self.p[1] - 0.05:
            return 2
        return 0

    def step(self, action: Any) -> BaseEnvTimestep:
        if self._vs_agent:
            a0, a1 = action[0], action[1]
        else:
            a0, a1 = action, self._bot()

        def as_int(a):
            return int(np.asarray(a).reshape(-1)[0])
        self._move(0, as_int(a0))
        self._move(1, as_int(a1))
        b = self.ball
        b[0] += b[2] * DT
        b[1] += b[3] * DT
        if abs(b[1]) > 1.0:  # top/bottom bounce
            b[3] = -b[3]
            b[1] = float(np.clip(b[1], -1.0, 1.0))
        reward = 0.0
        done = False
        for side, x_edge, direction in ((0, -1.0, 1), (1, 1.0, -1)):
            if (b[0] < x_edge if side == 0 else b[0] > x_edge):
                if abs(b[1] - self.p[side]) < 0.25:  # paddle hit: return ball
                    b[2] = -b[2] * 1.05
                    b[3] += self.pv[side] * 0.5
                    b[0] = x_edge + 0.01 * direction
                else:  # point against this side
                    other = 1 - side
                    self.score[other] += 1
                    reward = 1.0 if other == 0 else -1.0
                    self._serve(direction)
                if max(self.score) >= self._points:
                    done = True
        self._step_count += 1
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        if self._vs_agent:
            if done:
                info = {
                    'eval_episode_return': [self._eval_episode_return, -self._eval_episode_return],
                    'result': 'wins' if self.score[0] > self.score[1] else
                              ('losses' if self.score[0] < self.score[1] else 'draws'),
                }
            return BaseEnvTimestep([self._obs(0), self._obs(1)],
                                   np.array([reward, -reward], dtype=np.float32), done, info)
        return BaseEnvTimestep(self._obs(0), np.array([reward], dtype=np.float32), done, info)

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
        return "CompetitivePongEnv"
