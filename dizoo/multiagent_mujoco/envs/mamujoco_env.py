"""Multi-agent MuJoCo (reference dizoo/multiagent_mujoco/envs: factored
robots like 2-agent Hopper, each agent controls a subset of joints).
Offline implementation on the mujoco-lite smooth latent dynamics: the
action vector is split across agents; per-agent obs = global state + agent
one-hot; SMAC-style dict obs so continuous MARL policies (multi-agent SAC,
HAPPO-continuous) run unchanged.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnvTimestep
from ding.envs.common.spaces import Box
from ding.utils import ENV_REGISTRY
from dizoo.mujoco.envs.mujoco_lite_env import MujocoLiteEnv


@ENV_REGISTRY.register('mamujoco')
class MAMujocoEnv(MujocoLiteEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = dict(cfg or {})
        scenario = cfg.get('scenario', '2x3')  # agents x joints-per-agent
        n_agent, per = (int(x) for x in scenario.split('x'))
        cfg.setdefault('env_id', cfg.get('base_env', 'HalfCheetah-v3'))
        super().__init__(cfg)
        assert self.act_dim == n_agent * per, \
            f'scenario {scenario} does not factor action dim {self.act_dim}'
        self.agent_num = n_agent
        self.per = per
        self.agent_obs_dim = self.obs_dim + n_agent
        self._observation_space = Box(-np.inf, np.inf, (n_agent, self.agent_obs_dim))
        self._action_space = Box(-1.0, 1.0, (n_agent, per))

    def _ma_obs(self, state: np.ndarray) -> dict:
        per_agent = np.zeros((self.agent_num, self.agent_obs_dim), dtype=np.float32)
        for i in range(self.agent_num):
            one_hot = np.zeros(self.agent_num, dtype=np.float32)
            one_hot[i] = 1.0
            per_agent[i] = np.concatenate([state, one_hot])
        return {
            'agent_state': per_agent,
            'global_state': state.astype(np.float32),
            'action_mask': np.ones((self.agent_num, self.per), dtype=np.float32),
        }

    def reset(self) -> dict:
        return self._ma_obs(super().reset())

    def step(self, action: Any) -> BaseEnvTimestep:
        joint = np.asarray(action, dtype=np.float32).reshape(-1)[:self.act_dim]
        ts = super().step(joint)
        return BaseEnvTimestep(self._ma_obs(ts.obs), ts.reward, ts.done, ts.info)

    def random_action(self) -> np.ndarray:
        return self._rng.uniform(-1, 1, size=(self.agent_num, self.per)).astype(np.float32)

    def __repr__(self) -> str:
        return f"MAMujocoEnv({self.agent_num}x{self.per})"
