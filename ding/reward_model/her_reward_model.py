"""Hindsight experience replay goal-relabelling.

Parity: reference ding/reward_model/her_reward_model.py.
"""
import copy
import random
from typing import Any, Callable, List, Optional

import torch

from ding.utils import EasyDict


class HerRewardModel:
    """Relabel episode goals with achieved outcomes ('final'/'future'/
    'episode' strategies). The env must expose obs as a dict with
    'observation'/'achieved_goal'/'desired_goal' OR a user-supplied
    goal/reward function pair is used."""

    def __init__(self, cfg: dict = None, cuda: bool = False):
        cfg = EasyDict(cfg or {})
        self._cuda = cuda
        self._strategy = cfg.get('her_strategy', 'future')
        assert self._strategy in ('final', 'future', 'episode')
        self._replay_k = cfg.get('her_replay_k', 1)
        self._episode_size = cfg.get('episode_size', None)
        self._sample_per_episode = cfg.get('sample_per_episode', None)
        # user hooks: extract achieved goal from transition; compute reward
        self._get_goal: Callable = cfg.get('goal_fn', None) or (lambda t: t['obs'])
        self._reward_fn: Callable = cfg.get('reward_fn', None) or (
            lambda goal, t: torch.ones_like(t['reward']) if torch.allclose(
                torch.as_tensor(self._get_goal(t), dtype=torch.float32),
                torch.as_tensor(goal, dtype=torch.float32), atol=1e-2
            ) else torch.zeros_like(t['reward'])
        )

    @property
    def episode_size(self) -> Optional[int]:
        return self._episode_size

    @property
    def sample_per_episode(self) -> Optional[int]:
        return self._sample_per_episode

    def estimate(self, episode: List[dict]) -> List[List[dict]]:
        """Return relabelled copies of the episode (one per replay_k)."""
        out = []
        for _ in range(self._replay_k):
            new_episode = []
            for idx, t in enumerate(episode):
                nt = {k: (v.clone() if isinstance(v, torch.Tensor) else copy.deepcopy(v)) for k, v in t.items()}
                if self._strategy == 'final':
                    goal = self._get_goal(episode[-1])
                elif self._strategy == 'episode':
                    goal = self._get_goal(random.choice(episode))
                else:  # future
                    goal = self._get_goal(episode[random.randint(idx, len(episode) - 1)])
                nt['reward'] = self._reward_fn(goal, t)
                new_episode.append(nt)
            out.append(new_episode)
        return out
