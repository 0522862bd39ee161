"""Collector middleware classes.

Parity: reference ding/framework/middleware/collector.py (StepCollector:14,
EpisodeCollector:71, PPOFStepCollector:142).
"""
from typing import Optional

from ding.envs import BaseEnvManager
from ding.policy import Policy
from ding.utils import EasyDict
from ..context import OnlineRLContext
from .functional.collector import TransitionList, inferencer, rolloutor


class StepCollector:
    """Collect at least cfg.policy.collect.n_sample transitions per call
    (plus random warmup before random_collect_size env steps)."""

    def __init__(self, cfg: EasyDict, policy: Policy, env: BaseEnvManager, random_collect_size: int = 0) -> None:
        self.cfg = cfg
        self.env = env
        self.policy = policy
        self.random_collect_size = random_collect_size
        self._transitions = TransitionList(self.env.env_num)
        self._inferencer = inferencer(cfg.seed, policy, env)
        self._rolloutor = rolloutor(policy, env, self._transitions)

    def __call__(self, ctx: OnlineRLContext) -> None:
        old = ctx.env_step
        if self.random_collect_size > 0 and old < self.random_collect_size:
            target_size = self.random_collect_size - old
            random_policy = get_random_policy(self.cfg, self.policy, self.env)
            current_inferencer = inferencer(self.cfg.seed, random_policy, self.env)
        else:
            target_size = self.cfg.policy.collect.n_sample * self.cfg.policy.collect.unroll_len
            current_inferencer = self._inferencer

        while True:
            current_inferencer(ctx)
            self._rolloutor(ctx)
            if ctx.env_step - old >= target_size:
                ctx.trajectories, ctx.trajectory_end_idx = self._transitions.to_trajectories()
                self._transitions.clear()
                break


class EpisodeCollector:
    """Collect cfg.policy.collect.n_episode whole episodes per call."""

    def __init__(self, cfg: EasyDict, policy: Policy, env: BaseEnvManager, random_collect_size: int = 0) -> None:
        self.cfg = cfg
        self.env = env
        self.policy = policy
        self.random_collect_size = random_collect_size
        self._transitions = TransitionList(self.env.env_num)
        self._inferencer = inferencer(cfg.seed, policy, env)
        self._rolloutor = rolloutor(policy, env, self._transitions)

    def __call__(self, ctx: OnlineRLContext) -> None:
        old = ctx.env_episode
        if self.random_collect_size > 0 and ctx.env_step < self.random_collect_size:
            target_size = self.random_collect_size - ctx.env_step
            random_policy = get_random_policy(self.cfg, self.policy, self.env)
            current_inferencer = inferencer(self.cfg.seed, random_policy, self.env)
        else:
            target_size = self.cfg.policy.collect.n_episode
            current_inferencer = self._inferencer

        while True:
            current_inferencer(ctx)
            self._rolloutor(ctx)
            if ctx.env_episode - old >= target_size:
                ctx.episodes = self._transitions.to_episodes()
                self._transitions.clear()
                break


def get_random_policy(cfg: EasyDict, policy: Policy, env: BaseEnvManager):
    """A forward-compatible random policy view reusing process_transition."""

    class _RandomView:

        def forward(self, obs, **kwargs):
            actions = env.random_action()
            import torch
            return {i: {'action': torch.as_tensor(a), 'logit': None, 'value': None} for i, a in actions.items()}

        def process_transition(self, obs, policy_output, timestep):
            return policy.process_transition(obs, policy_output, timestep)

        def reset(self, env_ids=None):
            pass

    if cfg.policy.get('random_collect', None) and cfg.policy.random_collect.get('use_policy', False):
        return policy
    return _RandomView()


class PPOFStepCollector:
    """Step collector for the simplified PPOF policy interface (policy.collect
    on stacked obs tensors; transitions assembled via policy.process_transition).

    Parity: reference ding/framework/middleware/collector.py PPOFStepCollector.
    """

    def __init__(self, seed: int, policy, env: BaseEnvManager, n_sample: int, unroll_len: int = 1) -> None:
        self._policy = policy
        self._env = env
        self._n_sample = n_sample
        if env.closed:
            env.launch()
        self._transitions = []

    def __call__(self, ctx) -> None:
        import torch
        collected = []
        while len(collected) < self._n_sample:
            obs = self._env.ready_obs
            if not isinstance(obs, dict) or not all(isinstance(k, int) for k in obs):
                ids = self._env.ready_obs_id
                obs = {i: obs[pos] for pos, i in enumerate(ids)}
            ids = sorted(obs.keys())
            stacked = torch.stack([torch.as_tensor(obs[i], dtype=torch.float32) for i in ids])
            out = self._policy.collect(stacked)
            actions = {i: out['action'][k].cpu().numpy() for k, i in enumerate(ids)}
            timesteps = self._env.step(actions)
            if not isinstance(timesteps, dict):
                timesteps = {ts.info['env_id']: ts for ts in timesteps}
            for k, i in enumerate(ids):
                if i not in timesteps:
                    continue
                ts = timesteps[i]
                per = {kk: (vv[k] if isinstance(vv, torch.Tensor) else vv) for kk, vv in out.items()}
                collected.append(self._policy.process_transition(stacked[k], per, ts))
        ctx.trajectories = collected
        ctx.env_step = getattr(ctx, 'env_step', 0) + len(collected)
