"""Serial evaluator.

Parity: reference ding/worker/collector/interaction_serial_evaluator.py
(InteractionSerialEvaluator:14, should_eval:162, eval:184).
"""
import copy
import os
from typing import Any, Callable, Dict, List, Optional, Tuple

import numpy as np
import torch

from ding.envs import BaseEnvManager
from ding.torch_utils import to_ndarray, to_tensor
from ding.utils import SERIAL_EVALUATOR_REGISTRY, EasyDict, build_logger, deep_merge_dicts
from ding.framework.middleware.functional.evaluator import VectorEvalMonitor


@SERIAL_EVALUATOR_REGISTRY.register('interaction')
class InteractionSerialEvaluator:

    config = dict(
        type='interaction',
        eval_freq=1000,
        render=dict(render_freq=-1, mode='train_iter'),
        figure_path=None,
        stop_value=float("inf"),
        n_episode=None,
    )

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))

    def __init__(
        self,
        cfg: EasyDict,
        env: BaseEnvManager = None,
        policy=None,
        tb_logger=None,
        exp_name: str = 'default_experiment',
        instance_name: str = 'evaluator',
    ):
        self._cfg = deep_merge_dicts(self.default_config(), cfg or EasyDict({}))
        self._exp_name = exp_name
        self._instance_name = instance_name
        self._logger, self._tb_logger = build_logger(
            os.path.join(exp_name, 'log', instance_name), instance_name, need_tb=False
        )
        if tb_logger is not None:
            self._tb_logger = tb_logger
        self._stop_value = self._cfg.stop_value
        self._end_flag = False
        self._last_eval_iter = -1
        self._max_episode_return = float("-inf")
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
        self._policy.reset()

    def reset(self, _policy=None, _env: Optional[BaseEnvManager] = None) -> None:
        if _env is not None:
            self.reset_env(_env)
        if _policy is not None:
            self.reset_policy(_policy)
        self._max_episode_return = float("-inf")
        self._last_eval_iter = -1
        self._end_flag = False

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

    def should_eval(self, train_iter: int) -> bool:
        if train_iter == self._last_eval_iter:
            return False
        if (train_iter - self._last_eval_iter) < self._cfg.eval_freq and train_iter != 0:
            return False
        self._last_eval_iter = train_iter
        return True

    def eval(
        self,
        save_ckpt_fn: Optional[Callable] = None,
        train_iter: int = -1,
        envstep: int = -1,
        n_episode: Optional[int] = None,
        force_render: bool = False,
        policy_kwargs: Optional[Dict] = None,
    ) -> Tuple[bool, Dict[str, List]]:
        if n_episode is None:
            n_episode = self._cfg.n_episode or self._env_num
        self._env.reset()
        self._policy.reset()
        monitor = VectorEvalMonitor(self._env_num, n_episode)
        # recurrent-filter policies (dreamer): thread latent state + resets
        _recurrent = str(getattr(self._policy.get_attribute('cfg'), 'type', '')).startswith('dreamer')
        _states, _resets = None, np.zeros(self._env_num)
        while not monitor.is_finished():
            obs = self._env.ready_obs
            if not isinstance(obs, dict) or not all(isinstance(k, int) for k in obs):
                ids = self._env.ready_obs_id
                obs = {i: obs[pos] for pos, i in enumerate(ids)}
            obs_t = {i: to_tensor(o, dtype=torch.float32) for i, o in obs.items()}
            if _recurrent:
                policy_output = self._policy.forward(obs_t, **(policy_kwargs or {}), reset=_resets, state=_states)
                _states = [policy_output[i]['state'] for i in sorted(policy_output)]
            else:
                policy_output = self._policy.forward(obs_t, **(policy_kwargs or {}))
            actions = {i: to_ndarray(out['action']) for i, out in policy_output.items()}
            timesteps = self._env.step(actions)
            if not isinstance(timesteps, dict):
                timesteps = {ts.info['env_id']: ts for ts in timesteps}
            for env_id, timestep in timesteps.items():
                _resets[env_id] = float(bool(timestep.done))
                if timestep.done:
                    self._policy.reset([env_id])
                    monitor.update_reward(env_id, timestep.info.get('eval_episode_return', 0.0))
        episode_return = monitor.get_episode_return()
        episode_return_mean = float(np.mean(episode_return))
        if self._logger:
            self._logger.info(
                f'[{self._instance_name}] train_iter({train_iter}) envstep({envstep}) '
                f'eval_return({episode_return_mean:.3f})'
            )
        if self._tb_logger is not None:
            self._tb_logger.add_scalar('evaluator/eval_episode_return_mean', episode_return_mean, train_iter)
        if episode_return_mean > self._max_episode_return:
            if save_ckpt_fn is not None:
                save_ckpt_fn('ckpt_best.pth.tar')
            self._max_episode_return = episode_return_mean
        stop_flag = episode_return_mean >= self._stop_value and train_iter > 0
        episode_info = {'eval_episode_return': episode_return, 'train_iter': train_iter, 'ckpt_name': 'iteration_{}.pth.tar'.format(train_iter)}
        return stop_flag, episode_info


def create_serial_evaluator(cfg: EasyDict, **kwargs):
    cfg = copy.deepcopy(cfg)
    eval_type = cfg.pop('type', 'interaction')
    return SERIAL_EVALUATOR_REGISTRY.build(eval_type, cfg=cfg, **kwargs)
