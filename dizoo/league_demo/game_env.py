"""Two-player matrix-game battle env for league/battle-collector pipelines.

Parity: reference dizoo/league_demo/game_env.py (GameEnv). Re-designed as a
repeated game: each episode runs ``repeat_count`` rounds of a 2x2 matrix
game (zero_sum or prisoner_dilemma); obs/reward/info are per-player lists,
done is shared — the contract the 1v1 battle collectors expect.
"""
from typing import Any, List

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

# payoff[a0][a1] -> (r0, r1)
_PAYOFFS = {
    'zero_sum': {(0, 0): (3, -3), (0, 1): (-2, 2), (1, 0): (-2, 2), (1, 1): (1, -1)},
    'prisoner_dilemma': {(0, 0): (-1, -1), (0, 1): (-20, 0), (1, 0): (0, -20), (1, 1): (-10, -10)},
}


@ENV_REGISTRY.register('league_demo_game')
class GameEnv(BaseEnv):

    def __init__(self, cfg: Any = None) -> None:
        if isinstance(cfg, str):  # GameEnv('zero_sum') shorthand
            cfg = {'game_type': cfg}
        cfg = cfg or {}
        self.game_type = cfg.get('game_type', 'prisoner_dilemma')
        assert self.game_type in _PAYOFFS, self.game_type
        self.repeat_count = cfg.get('repeat_count', 1)
        self._observation_space = Box(0, 1, (2, 2))
        self._action_space = Discrete(2)
        self._reward_space = Box(-20, 3, (1, ))
        self._round = 0
        self._returns = None

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        pass  # deterministic payoff game

    def reset(self) -> List[np.ndarray]:
        self._round = 0
        self._returns = np.zeros(2, dtype=np.float32)
        obs = np.eye(2, dtype=np.float32)
        return [obs[0:2], obs[[1, 0]]]  # per-player views

    def step(self, actions: List[int]) -> BaseEnvTimestep:
        a = tuple(int(np.asarray(x).item()) for x in actions)
        r0, r1 = _PAYOFFS[self.game_type][a]
        self._returns += (r0, r1)
        self._round += 1
        done = self._round >= self.repeat_count
        obs = np.eye(2, dtype=np.float32)
        rewards = [np.array([r0], dtype=np.float32), np.array([r1], dtype=np.float32)]
        if done:
            results = ('wins', 'losses') if self._returns[0] > self._returns[1] else \
                      ('losses', 'wins') if self._returns[0] < self._returns[1] else ('draws', 'draws')
            infos = tuple(
                {'result': results[i], 'eval_episode_return': float(self._returns[i])} for i in range(2)
            )
        else:
            infos = ({}, {})
        return BaseEnvTimestep([obs[0:2], obs[[1, 0]]], rewards, done, infos)

    def close(self) -> None:
        pass

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
        return "LeagueDemo GameEnv({})".format(self.game_type)
