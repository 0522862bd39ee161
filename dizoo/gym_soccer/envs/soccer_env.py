"""Half-field offense soccer with hybrid actions (reference
dizoo/gym_soccer wrapping HFO). Native 2D implementation: the agent dribbles
toward goal and shoots; hybrid action = (type in {DASH, TURN, SHOOT},
args = [dash power/turn angle, direction]); +1 on goal, small shaping on
ball progress. Obs 10.
"""
from typing import Any
import math

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('gym_soccer')
class SoccerEnv(BaseEnv):

    DASH, TURN, SHOOT = 0, 1, 2

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_step = self._cfg.get('max_step', 200)
        self._observation_space = Box(-np.inf, np.inf, (10, ))
        self._action_space = Discrete(3)
        self._args_space = Box(-1.0, 1.0, (2, ))
        self._reward_space = Box(-1.0, 1.0, (1, ))
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
            self._args_space.seed(seed)
        self.px, self.py = self._rng.uniform(-0.8, -0.4), self._rng.uniform(-0.4, 0.4)
        self.theta = 0.0
        self.vx = self.vy = 0.0
        self.ball = np.array([self.px + 0.05, self.py])
        self.ball_v = np.zeros(2)
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._prev_ball_x = float(self.ball[0])
        return self._obs()

    def _obs(self) -> np.ndarray:
        has_ball = float(np.hypot(*(self.ball - [self.px, self.py])) < 0.1)
        return np.array([
            self.px, self.py, self.vx, self.vy, math.cos(self.theta), math.sin(self.theta),
            self.ball[0], self.ball[1], has_ball, self._step_count / self._max_step
        ], dtype=np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, dict):
            a_type = int(np.asarray(action.get('action_type', action.get('type'))).reshape(-1)[0])
            a_args = np.clip(np.asarray(action.get('action_args', action.get('args')),
                                        dtype=np.float64).reshape(-1), -1, 1)
        else:
            a_type, a_args = int(np.asarray(action).reshape(-1)[0]), np.zeros(2)
        dt = 0.1
        has_ball = np.hypot(*(self.ball - [self.px, self.py])) < 0.1
        if a_type == self.DASH:
            power = (a_args[0] + 1) / 2
            self.vx += power * math.cos(self.theta) * dt * 3
            self.vy += power * math.sin(self.theta) * dt * 3
        elif a_type == self.TURN:
            self.theta += a_args[0] * math.pi / 4
        elif a_type == self.SHOOT and has_ball:
            direction = a_args[1] * 0.5
            self.ball_v = np.array([2.0 * math.cos(direction), 2.0 * math.sin(direction)])
        self.vx *= 0.9
        self.vy *= 0.9
        self.px = float(np.clip(self.px + self.vx * dt, -1, 1))
        self.py = float(np.clip(self.py + self.vy * dt, -1, 1))
        if has_ball and a_type != self.SHOOT:
            self.ball = np.array([self.px + 0.05 * math.cos(self.theta), self.py + 0.05 * math.sin(self.theta)])
        else:
            self.ball = self.ball + self.ball_v * dt
            self.ball_v *= 0.95
        self._step_count += 1
        reward = (float(self.ball[0]) - self._prev_ball_x) * 0.1
        self._prev_ball_x = float(self.ball[0])
        done = False
        if self.ball[0] >= 1.0 and abs(self.ball[1]) < 0.25:  # goal mouth
            reward += 1.0
            done = True
        elif abs(self.ball[1]) > 1.0 or self.ball[0] < -1.0 or self.ball[0] >= 1.0:
            reward -= 0.2
            done = True
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

    def random_action(self) -> dict:
        return {
            'action_type': np.array([self._action_space.sample()], dtype=np.int64),
            'action_args': self._args_space.sample().astype(np.float32),
        }

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
        return "SoccerEnv"
