"""Functional collect middleware: inferencer + rolloutor.

Parity: reference ding/framework/middleware/functional/collector.py
(TransitionList:15, inferencer:50, rolloutor:89).
"""
from typing import Callable, List, Optional

import torch

from ding.envs import BaseEnvManager
from ding.policy import Policy
from ding.torch_utils import to_ndarray
from ding.utils import EasyDict
from ...context import OnlineRLContext


class TransitionList:
    """Per-env transition accumulator with episode-boundary queries."""

    def __init__(self, env_num: int):
        self.env_num = env_num
        self._transitions = [[] for _ in range(env_num)]
        self._done_idx = [[] for _ in range(env_num)]

    def append(self, env_id: int, transition) -> None:
        self._transitions[env_id].append(transition)
        if transition.done:
            self._done_idx[env_id].append(len(self._transitions[env_id]))

    def to_trajectories(self):
        trajectories = []
        trajectory_end_idx = []
        for env_id in range(self.env_num):
            trajectories.extend(self._transitions[env_id])
            trajectory_end_idx.append(len(trajectories) - 1)
        return trajectories, trajectory_end_idx

    def to_episodes(self):
        episodes = []
        for env_id in range(self.env_num):
            last_idx = 0
            for done_idx in self._done_idx[env_id]:
                episodes.append(self._transitions[env_id][last_idx:done_idx])
                last_idx = done_idx
        return episodes

    def clear(self):
        for t in self._transitions:
            t.clear()
        for d in self._done_idx:
            d.clear()

    def length(self, env_id: int) -> int:
        return len(self._transitions[env_id])


def inferencer(seed: int, policy: Policy, env: BaseEnvManager) -> Callable:
    """Batch policy inference over ready_obs -> ctx.action/ctx.inference_output."""
    if env.closed:
        env.launch()

    def _inference(ctx: OnlineRLContext):
        if env.closed:
            env.launch()
        ready = env.ready_obs
        if isinstance(ready, dict) and all(isinstance(k, int) for k in ready.keys()):
            # V1 manager: {env_id: np obs}
            obs = {i: torch.as_tensor(o, dtype=torch.float32) for i, o in ready.items()}
        elif isinstance(ready, dict):
            # dict-obs V2 manager: {key: stacked tensor}
            ids = env.ready_obs_id
            obs = {i: {k: v[pos].float() for k, v in ready.items()} for pos, i in enumerate(ids)}
        else:
            ids = env.ready_obs_id
            obs = {i: ready[pos].float() for pos, i in enumerate(ids)}
        ctx.obs = obs
        inference_kwargs = {}
        if 'collect_kwargs' in ctx and ctx.collect_kwargs:
            inference_kwargs = dict(ctx.collect_kwargs)
        inference_output = policy.forward(obs, **inference_kwargs)
        ctx.inference_output = inference_output
        ctx.action = {i: to_ndarray(v['action']) for i, v in inference_output.items()}

    return _inference


def rolloutor(policy: Policy, env: BaseEnvManager, transitions: TransitionList,
              collect_print_freq: int = 100) -> Callable:
    """Step envs with ctx.action, build transitions via
    policy.process_transition, track env_step/env_episode."""
    env_episode_id = [_ for _ in range(env.env_num)]
    current_id = env.env_num

    def _rollout(ctx: OnlineRLContext):
        nonlocal current_id
        from ding.envs import BaseEnvTimestep
        from ding.torch_utils import to_tensor
        timesteps = env.step(ctx.action)
        items = list(timesteps.items()) if isinstance(timesteps, dict) \
            else [(ts.info['env_id'], ts) for ts in timesteps]
        ctx.env_step += len(items)
        items = [
            (env_id, BaseEnvTimestep(to_tensor(ts.obs), to_tensor(ts.reward), ts.done, ts.info))
            for env_id, ts in items
        ]
        for env_id, timestep in items:
            transition = policy.process_transition(ctx.obs[env_id], ctx.inference_output[env_id], timestep)
            transition = EasyDict(transition)
            transition.collect_train_iter = torch.as_tensor([ctx.train_iter])
            transition.env_data_id = torch.as_tensor([env_episode_id[env_id]])
            transitions.append(env_id, transition)
            if timestep.done:
                policy.reset([env_id])
                env_episode_id[env_id] = current_id
                current_id += 1
                ctx.env_episode += 1

    return _rollout
