"""Google-football academy scenarios ("gfootball-lite", reference
dizoo/gfootball wrapping the gfootball engine). Implements
academy_empty_goal_close on the native soccer physics with the reference's
interface: simple115-style padded state obs and the 19-action set (subset
meaningful: idle/8 directions/shot; the rest alias to idle).
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY
from dizoo.gym_soccer.envs.soccer_env import SoccerEnv

DIRS = {1: 3.14, 2: -2.36, 3: -1.57, 4: -0.79, 5: 0.0, 6: 0.79, 7: 1.57, 8: 2.36}


@ENV_REGISTRY.register('gfootball')
class GFootballAcademyEnv(SoccerEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = dict(cfg or {})
        super().__init__(cfg)
        self._observation_space = Box(-np.inf, np.inf, (115, ))
        self._action_space = Discrete(19)

    def _obs115(self, base: np.ndarray) -> np.ndarray:
        v = np.zeros(115, dtype=np.float32)
        v[:len(base)] = base
        return v

    def reset(self) -> np.ndarray:
        return self._obs115(super().reset())

    def step(self, action: Any) -> BaseEnvTimestep:
        a = int(np.asarray(action).reshape(-1)[0])
        if a in DIRS:
            # run in direction: turn toward it, then dash
            self.theta = DIRS[a]
            hybrid = {'action_type': 0, 'action_args': np.array([1.0, 0.0])}
        elif a == 12:  # shot
            hybrid = {'action_type': 2, 'action_args': np.array([0.0, 0.0])}
        else:
            hybrid = {'action_type': 0, 'action_args': np.array([-1.0, 0.0])}
        ts = super().step(hybrid)
        return BaseEnvTimestep(self._obs115(ts.obs), ts.reward, ts.done, ts.info)

    def random_action(self) -> np.ndarray:
        return np.array([self._action_space.sample()], dtype=np.int64)

    def __repr__(self) -> str:
        return "GFootballAcademyEnv"
