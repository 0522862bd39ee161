"""1v1 battle collectors: N policies act in the same vectorized env; per-env
obs/timestep fields are per-policy lists and training samples are routed to
per-policy output lists.

Parity: reference ding/worker/collector/battle_sample_serial_collector.py
('sample_1v1') and battle_episode_serial_collector.py ('episode_1v1').
"""
import os
from typing import Any, List, Optional, Tuple

import torch

from ding.envs import BaseEnvManager
from ding.torch_utils import to_ndarray, to_tensor
from ding.utils import EasyDict, SERIAL_COLLECTOR_REGISTRY, SERIAL_EVALUATOR_REGISTRY, build_logger, deep_merge_dicts

from .sample_serial_collector import ISerialCollector


@SERIAL_COLLECTOR_REGISTRY.register('sample_1v1')
class BattleSampleSerialCollector(ISerialCollector):
    """Collect ``n_sample`` train samples for EACH policy that has
    collect_data enabled. ``collect`` returns ``(return_data, return_info)``,
    both lists indexed by policy id."""

    config = dict(type='sample_1v1', deepcopy_obs=False, transform_obs=False, collect_print_freq=100)

    def __init__(
        self,
        cfg: EasyDict,
        env: BaseEnvManager = None,
        policy: Optional[List[Any]] = None,
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

    def reset_policy(self, _policy: Optional[List[Any]] = None) -> None:
        if _policy is not None:
            self._policy = list(_policy)
            self._policy_num = len(self._policy)
            self._policy_cfg = self._policy[0].get_attribute('cfg')
            # which policies produce training data (inactive historical
            # opponents only act)
            self._policy_collect_data = [
                getattr(p, 'collect_data', True) if not isinstance(p, tuple) else True for p in self._policy
            ]
        for p in self._policy:
            p.reset()

    def reset(self, _policy: Optional[List[Any]] = None, _env: Optional[BaseEnvManager] = None) -> None:
        if _env is not None:
            self.reset_env(_env)
        if _policy is not None:
            self.reset_policy(_policy)
        self._obs_pool = {}
        self._policy_output_pool = {}
        self._traj_buffer = {
            env_id: {p: [] for p in range(self._policy_num)} for env_id in range(self._env_num)
        }
        self._env_info = {env_id: {'time': 0., 'step': 0} for env_id in range(self._env_num)}
        self._total_envstep_count = 0
        self._total_episode_count = 0

    @property
    def envstep(self) -> int:
        return self._total_envstep_count

    @envstep.setter
    def envstep(self, value: int) -> None:
        self._total_envstep_count = value

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

    def _forward_all(self, obs: dict, policy_kwargs: dict) -> Tuple[dict, dict]:
        """obs: {env_id: [obs_p0, obs_p1, ...]} -> per-policy forward.
        Returns (policy_output: list per policy of {env_id: out}, actions:
        {env_id: [a_p0, a_p1, ...]})."""
        per_policy_obs = [
            {env_id: o[p] for env_id, o in obs.items()} for p in range(self._policy_num)
        ]
        policy_output = [
            pol.forward(per_policy_obs[p], **policy_kwargs) for p, pol in enumerate(self._policy)
        ]
        actions = {}
        for p, out_d in enumerate(policy_output):
            for env_id, out in out_d.items():
                actions.setdefault(env_id, []).append(to_ndarray(out['action']))
        return policy_output, actions

    def _per_policy_timestep(self, timestep, p: int):
        """Slice per-policy fields (obs/reward/info are lists; done is shared)."""
        fields = []
        for name, v in zip(timestep._fields, timestep):
            if isinstance(v, bool) or name == 'done':
                fields.append(v)
            elif isinstance(v, (list, tuple)) and len(v) == self._policy_num:
                fields.append(v[p])
            else:
                fields.append(v)
        return type(timestep)(*fields)

    def collect(
        self,
        n_sample: Optional[int] = None,
        train_iter: int = 0,
        drop_extra: bool = True,
        policy_kwargs: Optional[dict] = None,
    ) -> Tuple[List[List[Any]], List[List[Any]]]:
        if n_sample is None:
            n_sample = self._policy_cfg.collect.n_sample
        if policy_kwargs is None:
            policy_kwargs = {}
        collected = [0 for _ in range(self._policy_num)]
        return_data = [[] for _ in range(self._policy_num)]
        return_info = [[] for _ in range(self._policy_num)]

        while any(c < n_sample for p, c in enumerate(collected) if self._policy_collect_data[p]):
            obs = self._env.ready_obs
            if not isinstance(obs, dict) or not all(isinstance(k, int) for k in obs):
                ids = self._env.ready_obs_id
                obs = {i: obs[pos] for pos, i in enumerate(ids)}
            obs = {i: to_tensor(o, dtype=torch.float32) for i, o in obs.items()}
            policy_output, actions = self._forward_all(obs, policy_kwargs)
            self._obs_pool.update(obs)
            for env_id in actions:
                self._policy_output_pool[env_id] = [policy_output[p][env_id] for p in range(self._policy_num)]
            timesteps = self._env.step(actions)
            if not isinstance(timesteps, dict):
                timesteps = {ts.info['env_id']: ts for ts in timesteps}

            for env_id, timestep in timesteps.items():
                self._env_info[env_id]['step'] += 1
                self._total_envstep_count += 1
                for p in range(self._policy_num):
                    if not self._policy_collect_data[p]:
                        continue
                    p_ts = self._per_policy_timestep(timestep, p)
                    p_ts = p_ts._replace(obs=to_tensor(p_ts.obs), reward=to_tensor(p_ts.reward))
                    transition = self._policy[p].process_transition(
                        self._obs_pool[env_id][p], self._policy_output_pool[env_id][p], p_ts
                    )
                    transition = EasyDict(transition)
                    transition.collect_iter = train_iter
                    self._traj_buffer[env_id][p].append(transition)
                    if timestep.done:
                        train_sample = self._policy[p].get_train_sample(self._traj_buffer[env_id][p])
                        return_data[p].extend(train_sample)
                        collected[p] += len(train_sample)
                        self._traj_buffer[env_id][p] = []
                if timestep.done:
                    self._total_episode_count += 1
                    for p, pol in enumerate(self._policy):
                        pol.reset([env_id])
                        info = timestep.info[p] if isinstance(timestep.info, (list, tuple)) else timestep.info
                        return_info[p].append(info)

        if drop_extra:
            return_data = [r[:n_sample] for r in return_data]
        return return_data, return_info


@SERIAL_COLLECTOR_REGISTRY.register('episode_1v1')
class BattleEpisodeSerialCollector(BattleSampleSerialCollector):
    """Collect whole episodes per policy (for episodic buffers / league jobs)."""

    config = dict(type='episode_1v1', deepcopy_obs=False, transform_obs=False, collect_print_freq=100)

    def collect(
        self,
        n_episode: Optional[int] = None,
        train_iter: int = 0,
        policy_kwargs: Optional[dict] = None,
    ) -> Tuple[List[List[Any]], List[List[Any]]]:
        if n_episode is None:
            n_episode = self._policy_cfg.collect.get('n_episode', 1)
        if policy_kwargs is None:
            policy_kwargs = {}
        collected_episode = 0
        return_data = [[] for _ in range(self._policy_num)]
        return_info = [[] for _ in range(self._policy_num)]

        while collected_episode < n_episode:
            obs = self._env.ready_obs
            if not isinstance(obs, dict) or not all(isinstance(k, int) for k in obs):
                ids = self._env.ready_obs_id
                obs = {i: obs[pos] for pos, i in enumerate(ids)}
            obs = {i: to_tensor(o, dtype=torch.float32) for i, o in obs.items()}
            policy_output, actions = self._forward_all(obs, policy_kwargs)
            self._obs_pool.update(obs)
            for env_id in actions:
                self._policy_output_pool[env_id] = [policy_output[p][env_id] for p in range(self._policy_num)]
            timesteps = self._env.step(actions)
            if not isinstance(timesteps, dict):
                timesteps = {ts.info['env_id']: ts for ts in timesteps}

            for env_id, timestep in timesteps.items():
                self._total_envstep_count += 1
                for p in range(self._policy_num):
                    if not self._policy_collect_data[p]:
                        continue
                    p_ts = self._per_policy_timestep(timestep, p)
                    p_ts = p_ts._replace(obs=to_tensor(p_ts.obs), reward=to_tensor(p_ts.reward))
                    transition = self._policy[p].process_transition(
                        self._obs_pool[env_id][p], self._policy_output_pool[env_id][p], p_ts
                    )
                    transition = EasyDict(transition)
                    transition.collect_iter = train_iter
                    self._traj_buffer[env_id][p].append(transition)
                if timestep.done:
                    collected_episode += 1
                    self._total_episode_count += 1
                    for p, pol in enumerate(self._policy):
                        if self._policy_collect_data[p]:
                            return_data[p].append(list(self._traj_buffer[env_id][p]))
                            self._traj_buffer[env_id][p] = []
                        pol.reset([env_id])
                        info = timestep.info[p] if isinstance(timestep.info, (list, tuple)) else timestep.info
                        return_info[p].append(info)

        return return_data, return_info


@SERIAL_EVALUATOR_REGISTRY.register('battle_interaction')
class BattleInteractionSerialEvaluator:
    """Evaluate policy 0 of a battle env against fixed opponents; stop on
    player-0 mean return reaching stop_value.

    Parity: reference ding/worker/collector/
    battle_interaction_serial_evaluator.py ('battle_interaction':17).
    """

    config = dict(type='battle_interaction', eval_freq=50, n_episode=4, stop_value=1e9)

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(cls.config)

    def __init__(self, cfg, env=None, policy=None, tb_logger=None,
                 exp_name: str = 'default_experiment', instance_name: str = 'battle_evaluator'):
        self._cfg = deep_merge_dicts(self.default_config(), cfg or EasyDict({}))
        self._env = env
        self._policy = list(policy)
        self._policy_num = len(self._policy)
        self._stop_value = self._cfg.stop_value
        self._last_eval_iter = -1

    def reset(self, _policy=None, _env=None) -> None:
        if _env is not None:
            self._env = _env
        if _policy is not None:
            self._policy = list(_policy)
        for p in self._policy:
            p.reset()

    def close(self) -> None:
        if self._env is not None:
            self._env.close()

    def should_eval(self, train_iter: int) -> bool:
        if train_iter == self._last_eval_iter:
            return False
        if (train_iter - self._last_eval_iter) < self._cfg.eval_freq and train_iter != 0:
            return False
        self._last_eval_iter = train_iter
        return True

    def eval(self, save_ckpt_fn=None, train_iter: int = -1, envstep: int = -1, n_episode=None):
        import numpy as np
        n_episode = n_episode or self._cfg.n_episode
        if self._env.closed:
            self._env.launch()
        else:
            self._env.reset()
        for p in self._policy:
            p.reset()
        returns = []
        while len(returns) < n_episode:
            obs = self._env.ready_obs
            if not isinstance(obs, dict) or not all(isinstance(k, int) for k in obs):
                ids = self._env.ready_obs_id
                obs = {i: obs[pos] for pos, i in enumerate(ids)}
            obs = {i: to_tensor(o, dtype=torch.float32) for i, o in obs.items()}
            per_policy_obs = [
                {env_id: o[p] for env_id, o in obs.items()} for p in range(self._policy_num)
            ]
            with torch.no_grad():
                outs = [pol.forward(per_policy_obs[p]) for p, pol in enumerate(self._policy)]
            actions = {}
            for p, out_d in enumerate(outs):
                for env_id, out in out_d.items():
                    actions.setdefault(env_id, []).append(to_ndarray(out['action']))
            timesteps = self._env.step(actions)
            if not isinstance(timesteps, dict):
                timesteps = {ts.info['env_id']: ts for ts in timesteps}
            for env_id, ts in timesteps.items():
                if ts.done:
                    info0 = ts.info[0] if isinstance(ts.info, (list, tuple)) else ts.info
                    returns.append(float(info0.get('eval_episode_return', 0.0)))
                    for p in self._policy:
                        p.reset([env_id])
        mean_ret = float(np.mean(returns))
        stop = mean_ret >= self._stop_value and train_iter > 0
        if stop and save_ckpt_fn:
            save_ckpt_fn('ckpt_best.pth.tar')
        return stop, {'eval_episode_return': returns, 'train_iter': train_iter}
