"""Beer-distribution game (reference dizoo/beergame/envs wrapping the
beergame simulator): the agent runs ONE echelon of a 4-stage supply chain
(retailer .. factory); the other echelons follow base-stock heuristics.
Obs: [inventory, backlog, on-order, last demand, incoming shipment];
action: order quantity offset in {0..4} around the observed demand;
reward: -(holding + 2 x backlog) cost per week, 52-week episodes.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY


@ENV_REGISTRY.register('beergame')
class BeerGameEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.role = int(self._cfg.get('role', 0))  # 0 retailer .. 3 factory
        self.weeks = int(self._cfg.get('weeks', 52))
        self._observation_space = Box(-np.inf, np.inf, (5, ))
        self._action_space = Discrete(5)
        self._reward_space = Box(-np.inf, 0.0, (1, ))
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
        self.inv = np.full(4, 12.0)       # per-echelon inventory
        self.backlog = np.zeros(4)
        self.pipeline = [[4.0, 4.0] for _ in range(4)]  # 2-week shipping lag
        self.last_demand = 4.0
        self.week = 0
        self._eval_episode_return = 0.0
        return self._obs()

    def _obs(self) -> np.ndarray:
        r = self.role
        return np.array([
            self.inv[r], self.backlog[r], sum(self.pipeline[r]), self.last_demand,
            self.pipeline[r][0]
        ], dtype=np.float32) / 10.0

    def step(self, action: Any) -> BaseEnvTimestep:
        if hasattr(action, 'reshape'):
            action = int(np.asarray(action).reshape(-1)[0])
        demand = float(self._rng.poisson(4) if self.week > 4 else 4)
        self.last_demand = demand
        orders = np.zeros(4)
        for i in range(4):
            if i == self.role:
                orders[i] = demand + (int(action) - 2)  # offset policy
            else:
                # base-stock heuristic for the scripted echelons
                target = 12.0
                orders[i] = max(0.0, demand + 0.5 * (target - self.inv[i] + self.backlog[i]))
            orders[i] = max(0.0, orders[i])
        # receive pipeline, serve downstream demand, place upstream orders
        downstream = demand
        for i in range(4):
            arriving = self.pipeline[i].pop(0)
            self.inv[i] += arriving
            want = downstream + self.backlog[i]
            shipped = min(self.inv[i], want)
            self.inv[i] -= shipped
            self.backlog[i] = want - shipped
            downstream = orders[i]
            self.pipeline[i].append(orders[i])
        cost = self.inv[self.role] * 0.5 + self.backlog[self.role] * 2.0
        reward = -float(cost) / 10.0
        self.week += 1
        done = self.week >= self.weeks
        self._eval_episode_return += reward
        info = {'eval_episode_return': self._eval_episode_return} if done else {}
        return BaseEnvTimestep(self._obs(), np.array([reward], dtype=np.float32), done, info)

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
        return "BeerGameEnv"
