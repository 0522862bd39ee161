"""Evaluator middleware: vectorized n-episode evaluation.

Parity: reference ding/framework/middleware/functional/evaluator.py
(VectorEvalMonitor:41, interaction_evaluator:213).
"""
import logging
from collections import deque
from typing import Callable, Optional

import numpy as np
import torch

from ding.envs import BaseEnvManager
from ding.policy import Policy
from ding.torch_utils import to_ndarray, to_tensor
from ding.utils import EasyDict
from ...context import OnlineRLContext, OfflineRLContext

logger = logging.getLogger('ding')


class VectorEvalMonitor:
    """Track per-env episode returns until n episodes finish; cap episodes
    per env at ceil(n/env_num) for unbiased averaging."""

    def __init__(self, env_num: int, n_episode: int):
        assert n_episode >= env_num, f"n_episode({n_episode}) < env_num({env_num})"
        self._env_num = env_num
        self._n_episode = n_episode
        each = n_episode // env_num
        extra = n_episode % env_num
        self._each_env_episode = [each + (1 if i < extra else 0) for i in range(env_num)]
        self._reward = {i: deque(maxlen=n) for i, n in enumerate(self._each_env_episode)}
        self._info = {i: deque(maxlen=n) for i, n in enumerate(self._each_env_episode)}

    def is_finished(self) -> bool:
        return all(len(self._reward[i]) >= n for i, n in enumerate(self._each_env_episode))

    def update_reward(self, env_id, reward) -> None:
        if isinstance(reward, torch.Tensor):
            reward = reward.item()
        self._reward[int(env_id)].append(float(reward))

    def update_info(self, env_id, info) -> None:
        self._info[int(env_id)].append(info)

    def get_episode_return(self) -> list:
        return sum([list(v) for v in self._reward.values()], [])

    def get_current_episode(self) -> int:
        return sum(len(v) for v in self._reward.values())


def interaction_evaluator(cfg: EasyDict, policy: Policy, env: BaseEnvManager, render: bool = False) -> Callable:
    """Evaluate every cfg.policy.eval.evaluator.eval_freq train iters; set
    ctx.eval_value and the stop flag when stop_value is reached."""
    env.seed(cfg.seed, dynamic_seed=False)

    def _evaluate(ctx):
        if ctx.last_eval_iter != -1 and ctx.train_iter - ctx.last_eval_iter < cfg.policy.eval.evaluator.eval_freq:
            return
        if env.closed:
            env.launch()
        else:
            env.reset()
        policy.reset()
        eval_monitor = VectorEvalMonitor(env.env_num, cfg.env.n_evaluator_episode)

        while not eval_monitor.is_finished():
            ready = env.ready_obs
            if isinstance(ready, dict) and all(isinstance(k, int) for k in ready.keys()):
                obs = {i: torch.as_tensor(o, dtype=torch.float32) for i, o in ready.items()}
            else:
                ids = env.ready_obs_id
                obs = {i: ready[pos].float() for pos, i in enumerate(ids)}
            inference_output = policy.forward(obs)
            action = {i: to_ndarray(v['action']) for i, v in inference_output.items()}
            timesteps = env.step(action)
            items = list(timesteps.items()) if isinstance(timesteps, dict) \
                else [(ts.info['env_id'], ts) for ts in timesteps]
            for env_id, timestep in items:
                if timestep.done:
                    policy.reset([env_id])
                    reward = timestep.info.get('eval_episode_return', 0.0)
                    eval_monitor.update_reward(env_id, reward)
                    if 'episode_info' in timestep.info:
                        eval_monitor.update_info(env_id, timestep.info['episode_info'])
        episode_return = eval_monitor.get_episode_return()
        eval_value = float(np.mean(episode_return))
        stop_flag = eval_value >= cfg.env.stop_value and ctx.train_iter > 0
        logger.info(f"Evaluation: train iter({ctx.train_iter}), eval value({eval_value:.3f})")
        ctx.last_eval_iter = ctx.train_iter
        ctx.eval_value = eval_value
        ctx.last_eval_value = eval_value
        ctx.eval_output = {'episode_return': episode_return, 'reward': episode_return}
        if stop_flag:
            from ding.framework import task as _task
            _task.finish = True

    return _evaluate


def metric_evaluator(cfg: EasyDict, policy, dataloader, metric) -> Callable:
    """Supervised-metric evaluation over a dataset (IC / BC eval)."""

    def _evaluate(ctx):
        if ctx.last_eval_iter != -1 and ctx.train_iter - ctx.last_eval_iter < cfg.policy.eval.evaluator.eval_freq:
            return
        results = []
        for batch in dataloader:
            out = policy.forward(batch)
            results.append(metric(out, batch))
        avg = float(np.mean(results))
        ctx.last_eval_iter = ctx.train_iter
        ctx.eval_value = avg

    return _evaluate


def interaction_evaluator_ttorch(cfg, policy, env, render: bool = False):
    """Reference exposes a treetensor-typed evaluator variant
    (functional/evaluator.py interaction_evaluator_ttorch); this build's
    contexts are plain dict/tensor, so the one evaluator serves both —
    kept as a named alias for API parity."""
    return interaction_evaluator(cfg, policy, env, render=render)
