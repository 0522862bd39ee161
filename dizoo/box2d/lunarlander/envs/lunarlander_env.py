"""LunarLander (discrete + continuous) with self-contained 2D rigid-body
physics — Box2D is unavailable offline, so the lander is integrated directly:
gravity, main/side thrusters with torque coupling, leg ground contact and
the gym reward shaping (potential on distance/speed/angle, fuel costs,
+-100 terminal). Interface parity with the reference
dizoo/box2d/lunarlander/envs/lunarlander_env.py: obs 8
[x, y, vx, vy, angle, omega, leg1, leg2], Discrete(4) for 'LunarLander-v2'
(noop / left / main / right) or Box(2) [main, lateral] for
'LunarLanderContinuous-v2'; stop_value 200.
"""
import math
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('lunarlander')
class LunarLanderEnv(BaseEnv):

    FPS = 50.0

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        env_id = self._cfg.get('env_id', 'LunarLander-v2')
        self._continuous = 'Continuous' in env_id or self._cfg.get('continuous', False)
        self._max_step = self._cfg.get('max_step', 1000)
        self._act_scale = self._cfg.get('act_scale', False)
        self._observation_space = Box(-np.inf, np.inf, (8, ))
        if self._continuous:
            self._action_space = Box(-1.0, 1.0, (2, ))
        else:
            self._action_space = Discrete(4)
        self._reward_space = Box(-200.0, 200.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    # --------------------------------------------------------------- physics
    # units: x in [-1, 1] (pad at 0), y >= 0 is altitude above the pad plane
    GRAVITY = -0.60        # units/s^2
    MAIN_ACC = 1.30        # main engine acceleration along body-up
    SIDE_ACC = 0.18        # lateral engine acceleration
    SIDE_TORQUE = 3.0      # rad/s^2 from a side engine
    ANGLE_DAMP = 0.4       # aerodynamic-ish angular damping

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        self._x = self._rng.uniform(-0.1, 0.1)
        self._y = 1.40
        self._vx = self._rng.uniform(-0.3, 0.3)
        self._vy = self._rng.uniform(-0.1, 0.0)
        self._theta = self._rng.uniform(-0.1, 0.1)
        self._omega = self._rng.uniform(-0.1, 0.1)
        self._legs = [False, False]
        self._step_count = 0
        self._eval_episode_return = 0.0
        self._prev_shaping = self._shaping()
        return self._obs()

    def _obs(self) -> np.ndarray:
        return np.array([
            self._x, self._y, self._vx, self._vy, self._theta, self._omega,
            1.0 if self._legs[0] else 0.0, 1.0 if self._legs[1] else 0.0
        ], dtype=np.float32)

    def _shaping(self) -> float:
        return (
            -100.0 * math.sqrt(self._x ** 2 + self._y ** 2)
            - 100.0 * math.sqrt(self._vx ** 2 + self._vy ** 2)
            - 100.0 * abs(self._theta) + 10.0 * self._legs[0] + 10.0 * self._legs[1]
        )

    def step(self, action: Any) -> BaseEnvTimestep:
        if self._continuous:
            a = np.clip(np.asarray(action, dtype=np.float64).reshape(-1), -1, 1)
            # gym semantics: main fires if a[0] > 0 with throttle in [0.5, 1]
            main = 0.5 + 0.5 * a[0] if a[0] > 0 else 0.0
            side = a[1] if abs(a[1]) > 0.5 else 0.0
        else:
            if isinstance(action, np.ndarray):
                action = int(action.item())
            action = int(action)
            main = 1.0 if action == 2 else 0.0
            side = {0: 0.0, 1: -1.0, 2: 0.0, 3: 1.0}[action]

        dt = 1.0 / self.FPS
        # thrust along body-up (rotated by theta)
        ax = -math.sin(self._theta) * self.MAIN_ACC * main + math.cos(self._theta) * self.SIDE_ACC * side
        ay = math.cos(self._theta) * self.MAIN_ACC * main + math.sin(self._theta) * self.SIDE_ACC * side
        ay += self.GRAVITY
        # side engines sit below the CoM: firing right pushes left AND torques
        alpha = -self.SIDE_TORQUE * side - self.ANGLE_DAMP * self._omega
        # dispersion noise like Box2D's particle impulses
        ax += self._rng.uniform(-1, 1) * 0.01 * main
        alpha += self._rng.uniform(-1, 1) * 0.05 * main

        self._vx += ax * dt
        self._vy += ay * dt
        self._x += self._vx * dt
        self._y += self._vy * dt
        self._omega += alpha * dt
        self._theta += self._omega * dt
        self._step_count += 1

        # ground contact (flat terrain, pad spans |x| <= 0.2)
        on_ground = self._y <= 0.0
        over_pad = abs(self._x) <= 0.2
        if on_ground:
            self._y = 0.0
            self._legs = [True, True]
        else:
            self._legs = [False, False]

        shaping = self._shaping()
        reward = shaping - self._prev_shaping
        self._prev_shaping = shaping
        reward -= 0.30 * main + 0.03 * abs(side)

        done = False
        if on_ground:
            soft = math.sqrt(self._vx ** 2 + self._vy ** 2) < 0.25 and abs(self._theta) < 0.35
            if soft and over_pad:
                # settled on the pad: success once velocity has bled off
                if abs(self._vx) < 0.05 and abs(self._vy) < 0.05:
                    reward += 100.0
                    done = True
                else:
                    self._vy = 0.0
                    self._vx *= 0.5
                    self._omega *= 0.5
            else:
                reward -= 100.0
                done = True
        if abs(self._x) > 1.0 or self._y > 2.0:
            reward -= 100.0
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

    def random_action(self) -> np.ndarray:
        if self._continuous:
            return self._action_space.sample().astype(np.float32)
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
        return "LunarLanderEnv"
