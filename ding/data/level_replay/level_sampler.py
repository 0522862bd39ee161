"""Prioritized Level Replay (PLR): sample training levels (env seeds) by
learning-potential scores (policy entropy / confidence / GAE magnitude / TD
error) mixed with a staleness term.

Parity: reference ding/data/level_replay/level_sampler.py (LevelSampler:9).
"""
from typing import List, Optional

import numpy as np
import torch

from ding.utils import EasyDict, deep_merge_dicts


class LevelSampler:

    config = dict(
        strategy='policy_entropy',
        replay_schedule='fixed',
        score_transform='rank',
        temperature=1.0,
        eps=0.05,
        rho=0.2,
        nu=0.5,
        alpha=1.0,
        staleness_coef=0.1,
        staleness_transform='power',
        staleness_temperature=1.0,
    )

    def __init__(self, seeds: List[int], obs_space, action_space: int, num_actors: int, cfg: EasyDict):
        self.cfg = EasyDict(deep_merge_dicts(self.config, cfg or {}))
        self.obs_space = obs_space
        self.action_space = action_space
        self.strategy = self.cfg.strategy
        self.replay_schedule = self.cfg.replay_schedule
        self.score_transform = self.cfg.score_transform
        self.temperature = self.cfg.temperature
        self.eps = self.cfg.eps
        self.rho = self.cfg.rho          # min seen fraction before replaying
        self.nu = self.cfg.nu            # P(new level) under fixed schedule
        self.alpha = self.cfg.alpha      # score EWA factor
        self.staleness_coef = self.cfg.staleness_coef
        self.staleness_transform = self.cfg.staleness_transform
        self.staleness_temperature = self.cfg.staleness_temperature

        self.seeds = np.array(seeds, dtype=np.int64)
        self.seed2index = {int(s): i for i, s in enumerate(seeds)}
        n = len(seeds)
        self.unseen_seed_weights = np.ones(n)
        self.seed_scores = np.zeros(n)
        self.partial_seed_scores = np.zeros((num_actors, n), dtype=np.float32)
        self.partial_seed_steps = np.zeros((num_actors, n), dtype=np.int64)
        self.seed_staleness = np.zeros(n)
        self.next_seed_index = 0

    # -------------------------------------------------------------- scoring
    def _score(self, episode_logits: torch.Tensor, **kw) -> float:
        s = self.strategy
        if s == 'policy_entropy':
            n = self.action_space
            max_ent = -np.log(1. / n)
            return float((-torch.exp(episode_logits) * episode_logits).sum(-1).mean()) / max_ent
        if s == 'least_confidence':
            return float((1 - torch.exp(episode_logits.max(-1, keepdim=True)[0])).mean())
        if s == 'min_margin':
            top2 = torch.exp(episode_logits.topk(2, dim=-1)[0])
            return 1 - float((top2[:, 0] - top2[:, 1]).mean())
        if s == 'gae':
            return float(kw['adv'].mean())
        if s == 'value_l1':
            return float(kw['adv'].abs().mean())
        if s == 'one_step_td_error':
            r, v = kw['rewards'], kw['value']
            T = len(r)
            return float((r[:T - 1] + v[:T - 1] - v[1:T]).abs().mean())
        raise ValueError(f'unsupported strategy: {s}')

    def update_with_rollouts(self, train_data: dict, num_actors: int) -> None:
        """train_data: flat [num_actors*T] tensors {seed, logit, done, reward?,
        adv?, value?}; episodes are segmented by done per actor."""
        if self.strategy in ('random', 'sequential'):
            return
        total = train_data['reward'].shape[0]
        T = total // num_actors
        view = lambda k: train_data[k].reshape(num_actors, T).transpose(0, 1)
        seeds = view('seed')
        done = view('done')
        logits = train_data['logit'].reshape(num_actors, T, -1).transpose(0, 1)
        need_v = self.strategy in ('gae', 'value_l1', 'one_step_td_error')
        if need_v:
            rewards, adv, value = view('reward'), view('adv'), view('value')
        for a in range(num_actors):
            done_steps = done[:, a].nonzero()[:, 0]
            start = 0
            for t in done_steps.tolist():
                if start >= T:
                    break
                if t == 0:
                    continue
                self._apply(a, seeds, logits, start, t,
                            (rewards, adv, value) if need_v else None, final=True)
                start = t
            if start < T:
                self._apply(a, seeds, logits, start, T,
                            (rewards, adv, value) if need_v else None, final=False)
        # flush remaining partials as full updates
        nz = np.argwhere(self.partial_seed_scores != 0)
        for a, i in nz:
            self.update_seed_score(a, i, 0, 0)
        self.partial_seed_scores.fill(0)
        self.partial_seed_steps.fill(0)

    def _apply(self, actor, seeds, logits, start, end, rav, final):
        seed = int(seeds[start, actor].item())
        idx = self.seed2index[seed]
        ep_logits = torch.log_softmax(logits[start:end, actor], -1)
        kw = {}
        if rav is not None:
            r, adv, v = rav
            kw = {'rewards': r[start:end, actor], 'adv': adv[start:end, actor], 'value': v[start:end, actor]}
        score = self._score(ep_logits, **kw)
        if final:
            self.update_seed_score(actor, idx, score, end - start)
        else:
            self._partial_update(actor, idx, score, end - start)

    def update_seed_score(self, actor, idx, score, num_steps):
        score = self._partial_update(actor, idx, score, num_steps, done=True)
        self.unseen_seed_weights[idx] = 0.
        self.seed_scores[idx] = (1 - self.alpha) * self.seed_scores[idx] + self.alpha * score

    def _partial_update(self, actor, idx, score, num_steps, done=False):
        ps = self.partial_seed_scores[actor][idx]
        pn = self.partial_seed_steps[actor][idx]
        running = pn + num_steps
        merged = ps + (score - ps) * num_steps / float(max(running, 1))
        if done:
            self.partial_seed_scores[actor][idx] = 0.
            self.partial_seed_steps[actor][idx] = 0
        else:
            self.partial_seed_scores[actor][idx] = merged
            self.partial_seed_steps[actor][idx] = running
        return merged

    # ------------------------------------------------------------- sampling
    def sample(self, strategy: Optional[str] = None) -> int:
        strategy = strategy or self.strategy
        if strategy == 'random':
            return int(np.random.choice(self.seeds))
        if strategy == 'sequential':
            seed = self.seeds[self.next_seed_index]
            self.next_seed_index = (self.next_seed_index + 1) % len(self.seeds)
            return int(seed)
        num_unseen = (self.unseen_seed_weights > 0).sum()
        seen_frac = (len(self.seeds) - num_unseen) / len(self.seeds)
        if self.replay_schedule == 'fixed':
            if seen_frac >= self.rho and (np.random.rand() > self.nu or seen_frac >= 1.0):
                return self._sample_replay_level()
            return self._sample_unseen_level()
        if seen_frac >= self.rho and np.random.rand() < seen_frac:
            return self._sample_replay_level()
        return self._sample_unseen_level()

    def _sample_replay_level(self) -> int:
        w = self._sample_weights()
        if np.isclose(w.sum(), 0):
            w = np.ones_like(w) / len(w)
        w = w / w.sum()  # staleness mixing can leave the sum slightly off 1
        idx = np.random.choice(len(self.seeds), p=w)
        self._update_staleness(idx)
        return int(self.seeds[idx])

    def _sample_unseen_level(self) -> int:
        w = self.unseen_seed_weights / self.unseen_seed_weights.sum()
        idx = np.random.choice(len(self.seeds), p=w)
        self._update_staleness(idx)
        return int(self.seeds[idx])

    def _update_staleness(self, idx) -> None:
        if self.staleness_coef > 0:
            self.seed_staleness += 1
            self.seed_staleness[idx] = 0

    def _sample_weights(self) -> np.ndarray:
        w = self._transform(self.score_transform, self.temperature, self.seed_scores)
        w = w * (1 - self.unseen_seed_weights)
        z = w.sum()
        if z > 0:
            w = w / z
        if self.staleness_coef > 0:
            sw = self._transform(self.staleness_transform, self.staleness_temperature, self.seed_staleness)
            sw = sw * (1 - self.unseen_seed_weights)
            z = sw.sum()
            if z > 0:
                sw = sw / z
            w = (1 - self.staleness_coef) * w + self.staleness_coef * sw
        return w

    def _transform(self, transform, temperature, scores) -> np.ndarray:
        scores = np.asarray(scores, dtype=np.float64)
        if transform == 'rank':
            order = np.flip(scores.argsort())
            ranks = np.empty_like(order)
            ranks[order] = np.arange(len(order)) + 1
            return 1 / ranks ** (1. / temperature)
        if transform == 'power':
            eps = 0 if self.staleness_coef > 0 else 1e-3
            return (scores + eps) ** (1. / temperature)
        raise ValueError(f'unsupported transform: {transform}')
