"""gym-hybrid Moving-v0 / Sliding-v0 implemented natively.

Parity with the reference dizoo/gym_hybrid/envs/gym_hybrid_env.py wrapping
the gym-hybrid package: an agent on the unit disk must stop inside the
target circle. Hybrid action = (action_type in {ACCELERATE, TURN, BREAK},
action_args = [acceleration value, turn angle]); only the arg matching the
chosen type is applied (Moving) or both scaled (Sliding ignores turning
inertia). Observation 10-dim: [x, y, vx, vy, cos th, sin th, target_x,
target_y, distance, step_fraction]; reward = distance decrease - 0.001 step
penalty, terminal +1 when stopped in the zone (speed < 0.1), -1 out of
bounds; stop_value 1.8.
"""
from typing import Any
import math

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('gym_hybrid')
class MovingEnv(BaseEnv):

    ACCELERATE, TURN, BREAK = 0, 1, 2

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._env_id = self._cfg.get('env_id', 'Moving-v0')
        self._max_step = self._cfg.get('max_step', 200)
        self._act_scale = self._cfg.get('act_scale', True)
        self._target_radius = 0.1
        self._observation_space = Box(-np.inf, np.inf, (10, ))
        self._action_space = Discrete(3)  # plus Box(2) args
        self._args_space = Box(-1.0, 1.0, (2, ))
        self._reward_space = Box(-1.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._args_space.seed(seed)
        ang = self._rng.uniform(0, 2 * math.pi)
        r = self._rng.uniform(0.6, 0.95)
        self._pos = np.array([r * math.cos(ang), r * math.sin(ang)])
        self._vel = np.zeros(2)
        self._theta = self._rng.uniform(0, 2 * math.pi)
        self._target = np.zeros(2)
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._prev_dist = float(np.linalg.norm(self._pos - self._target))
        return self._obs()

    def _obs(self) -> np.ndarray:
        d = float(np.linalg.norm(self._pos - self._target))
        return np.array([
            self._pos[0], self._pos[1], self._vel[0], self._vel[1],
            math.cos(self._theta), math.sin(self._theta),
            self._target[0], self._target[1], d, self._step_count / self._max_step
        ], dtype=np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, dict):
            a_type = action.get('action_type', action.get('type'))
            a_args = action.get('action_args', action.get('args'))
        else:  # flat [type, accel, angle]
            a_type, a_args = action[0], action[1:]
        if isinstance(a_type, np.ndarray):
            a_type = int(a_type.reshape(-1)[0])
        a_type = int(a_type)
        a_args = np.clip(np.asarray(a_args, dtype=np.float64).reshape(-1), -1, 1)
        accel = float(a_args[0]) if len(a_args) > 0 else 0.0
        angle = float(a_args[1]) if len(a_args) > 1 else 0.0

        dt = 0.1
        if a_type == self.ACCELERATE:
            self._vel += accel * 0.5 * np.array([math.cos(self._theta), math.sin(self._theta)]) * dt
        elif a_type == self.TURN:
            self._theta += angle * math.pi / 3 * dt * 10
        elif a_type == self.BREAK:
            self._vel *= 0.2
        if 'Sliding' not in self._env_id:
            self._vel *= 0.95  # Moving has friction; Sliding is frictionless
        self._pos = self._pos + self._vel * dt
        self._step_count += 1

        d = float(np.linalg.norm(self._pos - self._target))
        reward = (self._prev_dist - d) - 0.001
        self._prev_dist = d
        done = False
        speed = float(np.linalg.norm(self._vel))
        if d < self._target_radius and speed < 0.1:
            reward += 1.0
            done = True
        elif float(np.linalg.norm(self._pos)) > 1.5:
            reward -= 1.0
            done = True
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

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
        return "MovingEnv"
