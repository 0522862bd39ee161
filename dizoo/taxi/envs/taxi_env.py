"""Taxi-v3 implemented natively (gymnasium unavailable offline).

Exact 5x5 map with the four depots R(0,0) G(0,4) Y(4,0) B(4,3) and the
classic wall layout; actions south/north/east/west/pickup/dropoff; rewards
-1 per step, +20 successful dropoff, -10 illegal pickup/dropoff. Observation
is the reference's 34-dim encoding (25 one-hot taxi cell + 5 passenger
location incl. in-taxi + 4 destination), parity with
dizoo/taxi/envs/taxi_env.py (_encode_taxi) and taxi_dqn_config.py
(obs_shape=34, action_shape=6, stop_value=20).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

# depots in (row, col): R, G, Y, B
DEPOTS = [(0, 0), (0, 4), (4, 0), (4, 3)]
# vertical walls between (row, col) and (row, col+1), from the gym map
WALLS = {(0, 1), (1, 1), (3, 0), (4, 0), (3, 2), (4, 2)}


@ENV_REGISTRY.register('taxi')
class TaxiEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._max_step = self._cfg.get('max_step', 200)
        self._observation_space = Box(0.0, 1.0, (34, ))
        self._action_space = Discrete(6)
        self._reward_space = Box(-10.0, 20.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def reset(self) -> np.ndarray:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
            self._action_space.seed(seed)
        self._row, self._col = self._rng.randint(0, 5), self._rng.randint(0, 5)
        self._passenger = self._rng.randint(0, 4)  # depot index; 4 = in taxi
        self._dest = self._rng.randint(0, 4)
        while self._dest == self._passenger:
            self._dest = self._rng.randint(0, 4)
        self._step_count = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        v = np.zeros(34, dtype=np.float32)
        v[5 * self._row + self._col] = 1.0
        v[25 + self._passenger] = 1.0
        v[30 + self._dest] = 1.0
        return v

    def step(self, action: Any) -> BaseEnvTimestep:
        if isinstance(action, np.ndarray):
            action = int(action.item())
        action = int(action)
        reward = -1.0
        done = False
        if action == 0:  # south
            self._row = min(self._row + 1, 4)
        elif action == 1:  # north
            self._row = max(self._row - 1, 0)
        elif action == 2:  # east
            if (self._row, self._col) not in WALLS:
                self._col = min(self._col + 1, 4)
        elif action == 3:  # west
            if (self._row, self._col - 1) not in WALLS:
                self._col = max(self._col - 1, 0)
        elif action == 4:  # pickup
            if self._passenger < 4 and (self._row, self._col) == DEPOTS[self._passenger]:
                self._passenger = 4
            else:
                reward = -10.0
        elif action == 5:  # dropoff
            if self._passenger == 4 and (self._row, self._col) == DEPOTS[self._dest]:
                reward = 20.0
                done = True
            else:
                reward = -10.0
        self._step_count += 1
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
        return "TaxiEnv"
