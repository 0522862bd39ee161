"""BipedalWalker-shaped continuous-control env (Box2D unavailable offline).

Preserves the reference interface (dizoo/box2d/bipedalwalker/envs/
bipedalwalker_env.py): obs 24 (hull angle/velocities, 4 joint angles +
speeds, leg contacts, 10 lidar), Box(4) torques, episode 1600, reward =
forward progress - torque cost, fall penalty -100; stop_value 300. The
hidden dynamics are a smooth articulated-chain surrogate: joint torques
drive leg phase oscillators whose symmetry determines hull speed and
stability, so policies must learn coordinated, low-torque gaits.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('bipedalwalker')
class BipedalWalkerEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_step = self._cfg.get('max_step', 1600)
        self._act_scale = self._cfg.get('act_scale', True)
        self._observation_space = Box(-np.inf, np.inf, (24, ))
        self._action_space = Box(-1.0, 1.0, (4, ))
        self._reward_space = Box(-100.0, 100.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        self._hull_angle = self._rng.uniform(-0.05, 0.05)
        self._hull_omega = 0.0
        self._vx = 0.0
        self._joints = self._rng.uniform(-0.1, 0.1, size=4)   # hip1, knee1, hip2, knee2
        self._joint_vel = np.zeros(4)
        self._x = 0.0
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        contacts = [1.0 if self._joints[1] < 0 else 0.0, 1.0 if self._joints[3] < 0 else 0.0]
        lidar = np.full(10, 1.0)  # flat terrain
        return np.concatenate([
            [self._hull_angle, self._hull_omega, self._vx, 0.0],
            np.stack([self._joints[0], self._joint_vel[0], self._joints[1], self._joint_vel[1]]),
            [contacts[0]],
            np.stack([self._joints[2], self._joint_vel[2], self._joints[3], self._joint_vel[3]]),
            [contacts[1]],
            lidar,
        ]).astype(np.float32)

    def step(self, action: Any) -> BaseEnvTimestep:
        a = np.clip(np.asarray(action, dtype=np.float64).reshape(-1), -1, 1)
        dt = 1.0 / 50.0
        # joint dynamics: torque-driven, spring-damped around neutral
        self._joint_vel += (4.0 * a - 2.0 * self._joints - 0.8 * self._joint_vel) * dt * 5.0
        self._joints = np.clip(self._joints + self._joint_vel * dt, -1.5, 1.5)
        # anti-phase leg coordination propels the hull; co-phase destabilizes
        stride = (self._joints[0] - self._joints[2]) * (self._joint_vel[0] - self._joint_vel[2])
        self._vx += (0.8 * np.clip(stride, -1, 2) - 0.3 * self._vx) * dt * 5.0
        # hull stability reacts to asymmetric knee loading
        self._hull_omega += ((self._joints[1] + self._joints[3]) * 0.5 - 1.5 * self._hull_angle
                             - 0.5 * self._hull_omega) * dt * 5.0
        self._hull_angle += self._hull_omega * dt * 5.0
        self._x += max(self._vx, -1.0) * dt
        self._step_count += 1

        reward = 130.0 * max(self._vx, -1.0) * dt - 0.001 * float(np.abs(a).sum()) \
            - 5.0 * abs(self._hull_angle) * dt
        done = self._step_count >= self._max_step
        if abs(self._hull_angle) > 0.8:  # fell over
            reward = -100.0
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
        return "BipedalWalkerEnv"
