"""Serial sample collector: drives the env manager with policy collect_mode
until n_sample train samples are gathered.

Parity: reference ding/worker/collector/sample_serial_collector.py
(SampleSerialCollector, hot loop :224-290).
"""
import copy
import os
from collections import namedtuple
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from ding.envs import BaseEnvManager
from ding.torch_utils import to_ndarray, to_tensor
from ding.utils import SERIAL_COLLECTOR_REGISTRY, EasyDict, build_logger, deep_merge_dicts, one_time_warning


class ISerialCollector:

    config = dict()

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))


@SERIAL_COLLECTOR_REGISTRY.register('sample')
class SampleSerialCollector(ISerialCollector):

    config = dict(type='sample', deepcopy_obs=False, transform_obs=False, collect_print_freq=100)

    def __init__(
        self,
        cfg: EasyDict,
        env: BaseEnvManager = None,
        policy=None,
        tb_logger=None,
        exp_name: str = 'default_experiment',
        instance_name: str = 'collector',
    ):
        self._cfg = deep_merge_dicts(self.default_config(), cfg or EasyDict({}))
        self._exp_name = exp_name
        self._instance_name = instance_name
        self._logger, self._tb_logger = build_logger(
            os.path.join(exp_name, 'log', instance_name), instance_name, need_tb=False
        )
        if tb_logger is not None:
            self._tb_logger = tb_logger
        self._end_flag = False
        self._env = None
        self._policy = None
        self.reset(policy, env)

    def reset_env(self, _env: Optional[BaseEnvManager] = None) -> None:
        if _env is not None:
            self._env = _env
            if self._env.closed:
                self._env.launch()
            else:
                self._env.reset()
            self._env_num = self._env.env_num
        else:
            self._env.reset()

    def reset_policy(self, _policy=None) -> None:
        if _policy is not None:
            self._policy = _policy
            self._policy_cfg = self._policy.get_attribute('cfg')
            self._unroll_len = self._policy.get_attribute('unroll_len')
            self._on_policy = self._policy.get_attribute('cfg').on_policy
        self._policy.reset()

    def reset(self, _policy=None, _env: Optional[BaseEnvManager] = None) -> None:
        if _env is not None:
            self.reset_env(_env)
        if _policy is not None:
            self.reset_policy(_policy)
        self._env_info = {env_id: {'time': 0., 'step': 0} for env_id in range(self._env_num)}
        self._obs_pool = {}
        self._policy_output_pool = {}
        self._traj_buffer = {env_id: [] for env_id in range(self._env_num)}
        # recurrent-filter policies (dreamer): per-env latent state + reset flags
        self._states = None
        self._resets = __import__('numpy').zeros(self._env_num)
        self._total_envstep_count = 0
        self._total_episode_count = 0
        self._total_train_sample_count = 0
        self._last_train_iter = 0

    @property
    def envstep(self) -> int:
        return self._total_envstep_count

    def close(self) -> None:
        if self._end_flag:
            return
        self._end_flag = True
        if self._env is not None:
            self._env.close()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def collect(
        self,
        n_sample: Optional[int] = None,
        train_iter: int = 0,
        drop_extra: bool = True,
        random_collect: bool = False,
        record_random_collect: bool = True,
        policy_kwargs: Optional[dict] = None,
        level_seeds: Optional[List] = None,
    ) -> List[Any]:
        if n_sample is None:
            n_sample = self._policy_cfg.collect.n_sample
        if policy_kwargs is None:
            policy_kwargs = {}
        collected_sample = 0
        return_data = []

        while collected_sample < n_sample:
            obs = self._env.ready_obs
            if not isinstance(obs, dict) or not all(isinstance(k, int) for k in obs):
                ids = self._env.ready_obs_id
                obs = {i: obs[pos] for pos, i in enumerate(ids)}
            obs_t = {i: to_tensor(o, dtype=torch.float32) for i, o in obs.items()}
            if random_collect:
                actions = self._env.random_action()
                policy_output = {i: {'action': torch.as_tensor(a)} for i, a in actions.items()}
            elif str(self._policy_cfg.type).startswith('dreamer'):
                # thread the RSSM filter state through the policy
                policy_output = self._policy.forward(
                    obs_t, **policy_kwargs, reset=self._resets, state=self._states
                )
                self._states = [policy_output[i]['state'] for i in sorted(policy_output)]
            else:
                policy_output = self._policy.forward(obs_t, **policy_kwargs)
            self._obs_pool = obs_t
            self._policy_output_pool = policy_output
            actions = {i: to_ndarray(out['action']) for i, out in policy_output.items()}
            timesteps = self._env.step(actions)
            if not isinstance(timesteps, dict):
                timesteps = {ts.info['env_id']: ts for ts in timesteps}

            for env_id, timestep in timesteps.items():
                if timestep.info.get('abnormal', False):
                    self._env.reset({env_id: {}})
                    self._policy.reset([env_id])
                    self._traj_buffer[env_id].clear()
                    continue
                ts = timestep._replace(obs=to_tensor(timestep.obs), reward=to_tensor(timestep.reward))
                transition = self._policy.process_transition(self._obs_pool[env_id], policy_output[env_id], ts)
                transition = EasyDict(transition)
                transition.collect_iter = train_iter
                if level_seeds is not None:
                    # PLR: tag each transition with its env's level seed
                    transition.seed = level_seeds[env_id]
                self._resets[env_id] = float(bool(timestep.done))
                self._traj_buffer[env_id].append(transition)
                self._env_info[env_id]['step'] += 1
                self._total_envstep_count += 1

                if timestep.done:
                    transitions = self._traj_buffer[env_id]
                    train_sample = self._policy.get_train_sample(transitions)
                    return_data.extend(train_sample)
                    collected_sample += len(train_sample)
                    self._traj_buffer[env_id] = []
                    self._total_episode_count += 1
                    self._policy.reset([env_id])

            # off-policy flush without waiting for done (traj_len_inf False)
            if True:
                for env_id in list(self._traj_buffer.keys()):
                    buf = self._traj_buffer[env_id]
                    flush_len = self._get_flush_len()
                    if flush_len is not None and len(buf) >= flush_len:
                        train_sample = self._policy.get_train_sample(buf)
                        return_data.extend(train_sample)
                        collected_sample += len(train_sample)
                        self._traj_buffer[env_id] = []

        self._total_train_sample_count += len(return_data)
        if drop_extra and len(return_data) > n_sample:
            return_data = return_data[:n_sample]
        return return_data

    def _get_flush_len(self) -> Optional[int]:
        """Flush segment length for non-episodic off-policy collection:
        enough steps that n-step return and unroll splitting stay valid."""
        cfg = self._policy_cfg
        if cfg.on_policy:
            return None
        nstep = cfg.get('nstep', 1) or 1
        unroll = self._unroll_len or 1
        return max(32, nstep * 2, unroll)


@SERIAL_COLLECTOR_REGISTRY.register('episode')
class EpisodeSerialCollector(SampleSerialCollector):

    config = dict(type='episode', deepcopy_obs=False, transform_obs=False, collect_print_freq=100,
                  get_train_sample=False, reward_shaping=False)

    def collect(self, n_episode: Optional[int] = None, train_iter: int = 0, policy_kwargs: Optional[dict] = None,
                **kwargs) -> List[Any]:
        if n_episode is None:
            n_episode = self._policy_cfg.collect.n_episode
        if policy_kwargs is None:
            policy_kwargs = {}
        episodes = []
        while len(episodes) < n_episode:
            obs = self._env.ready_obs
            if not isinstance(obs, dict) or not all(isinstance(k, int) for k in obs):
                ids = self._env.ready_obs_id
                obs = {i: obs[pos] for pos, i in enumerate(ids)}
            obs_t = {i: to_tensor(o, dtype=torch.float32) for i, o in obs.items()}
            policy_output = self._policy.forward(obs_t, **policy_kwargs)
            actions = {i: to_ndarray(out['action']) for i, out in policy_output.items()}
            timesteps = self._env.step(actions)
            if not isinstance(timesteps, dict):
                timesteps = {ts.info['env_id']: ts for ts in timesteps}
            for env_id, timestep in timesteps.items():
                ts = timestep._replace(obs=to_tensor(timestep.obs), reward=to_tensor(timestep.reward))
                transition = self._policy.process_transition(obs_t[env_id], policy_output[env_id], ts)
                transition = EasyDict(transition)
                transition.collect_iter = train_iter
                self._traj_buffer[env_id].append(transition)
                self._total_envstep_count += 1
                if timestep.done:
                    if self._cfg.get('get_train_sample', False):
                        episodes.append(self._policy.get_train_sample(self._traj_buffer[env_id]))
                    else:
                        episodes.append(self._traj_buffer[env_id])
                    self._traj_buffer[env_id] = []
                    self._total_episode_count += 1
                    self._policy.reset([env_id])
        return episodes


def create_serial_collector(cfg: EasyDict, **kwargs):
    cfg = copy.deepcopy(cfg)
    collector_type = cfg.pop('type', 'sample')
    return SERIAL_COLLECTOR_REGISTRY.build(collector_type, cfg=cfg, **kwargs)


def get_serial_collector_cls(cfg: EasyDict) -> type:
    return SERIAL_COLLECTOR_REGISTRY.get(cfg.get('type', 'sample'))
