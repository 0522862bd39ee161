"""Legacy replay buffers: naive ring buffer and PER advanced buffer.

Parity: reference ding/worker/replay_buffer/naive_buffer.py:15 and
advanced_buffer.py:23 (sum/min tree PER with IS weights, staleness and
use-count eviction). The PER hot path runs on the C++ segment tree
(ding/utils/_ctree).
"""
import copy
import time
from typing import Any, List, Optional, Union

import numpy as np

from ding.utils import BUFFER_REGISTRY, EasyDict, SumSegmentTree, MinSegmentTree, LockContext, LockContextType, \
    deep_merge_dicts, build_logger


class IBuffer:

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))


@BUFFER_REGISTRY.register('naive')
class NaiveReplayBuffer(IBuffer):
    """Thread-safe FIFO ring buffer with uniform sampling."""

    config = dict(
        type='naive',
        replay_buffer_size=10000,
        deepcopy=False,
        enable_track_used_data=False,
        periodic_thruput_seconds=60,
    )

    def __init__(self, cfg: EasyDict, tb_logger=None, exp_name: str = 'default_experiment',
                 instance_name: str = 'buffer'):
        self._cfg = deep_merge_dicts(EasyDict(copy.deepcopy(self.config)), cfg or EasyDict({}))
        self._exp_name = exp_name
        self._instance_name = instance_name
        self._replay_buffer_size = self._cfg.replay_buffer_size
        self._deepcopy = self._cfg.deepcopy
        self._data: List[Any] = [None] * self._replay_buffer_size
        self._valid_count = 0
        self._tail = 0
        self._push_count = 0
        self._lock = LockContext(LockContextType.THREAD_LOCK)

    def start(self) -> None:
        pass

    def close(self) -> None:
        pass

    def push(self, data: Union[List[Any], Any], cur_collector_envstep: int = -1) -> None:
        if isinstance(data, list):
            for d in data:
                self._push(d)
        else:
            self._push(data)

    def _push(self, data: Any) -> None:
        with self._lock:
            if self._deepcopy:
                data = copy.deepcopy(data)
            self._data[self._tail] = data
            self._tail = (self._tail + 1) % self._replay_buffer_size
            self._valid_count = min(self._valid_count + 1, self._replay_buffer_size)
            self._push_count += 1

    def sample(self, size: int, cur_learner_iter: int = -1, sample_range=None) -> Optional[List[Any]]:
        if size == 0:
            return []
        with self._lock:
            if self._valid_count < size:
                return None
            indices = np.random.choice(self._valid_count, size, replace=False)
            return [self._data[i] for i in indices]

    def update(self, info: dict) -> None:
        pass

    def clear(self) -> None:
        with self._lock:
            self._data = [None] * self._replay_buffer_size
            self._valid_count = 0
            self._tail = 0

    def count(self) -> int:
        return self._valid_count

    @property
    def replay_buffer_size(self):
        return self._replay_buffer_size

    @property
    def push_count(self):
        return self._push_count

    def state_dict(self) -> dict:
        return {
            'data': self._data, 'tail': self._tail, 'valid_count': self._valid_count,
            'push_count': self._push_count
        }

    def load_state_dict(self, d: dict) -> None:
        self._data = d['data']
        self._tail = d['tail']
        self._valid_count = d['valid_count']
        self._push_count = d['push_count']


@BUFFER_REGISTRY.register('advanced')
class AdvancedReplayBuffer(NaiveReplayBuffer):
    """PER with IS weights, use-count and staleness eviction."""

    config = dict(
        type='advanced',
        replay_buffer_size=4096,
        max_use=float("inf"),
        max_staleness=float("inf"),
        alpha=0.6,
        beta=0.4,
        anneal_step=int(1e5),
        enable_track_used_data=False,
        deepcopy=False,
        thruput_controller=dict(push_sample_rate_limit=dict(max=float("inf"), min=0), window_seconds=30),
        monitor=dict(),
    )

    def __init__(self, cfg: EasyDict, tb_logger=None, exp_name: str = 'default_experiment',
                 instance_name: str = 'buffer'):
        super().__init__(cfg, tb_logger, exp_name, instance_name)
        self._max_use = self._cfg.max_use
        self._max_staleness = self._cfg.max_staleness
        self.alpha = self._cfg.alpha
        self._beta = self._cfg.beta
        self._anneal_step = self._cfg.anneal_step
        if self._anneal_step != 0:
            self._beta_anneal_step = (1 - self._beta) / self._anneal_step
        capacity = int(np.power(2, np.ceil(np.log2(self._replay_buffer_size))))
        self._sum_tree = SumSegmentTree(capacity)
        self._min_tree = MinSegmentTree(capacity)
        self._max_priority = 1.0
        self._eps = 1e-5

    def _push(self, data: Any) -> None:
        with self._lock:
            if self._deepcopy:
                data = copy.deepcopy(data)
            if isinstance(data, dict):
                data.setdefault('priority', None)
                data['replay_unique_id'] = self._push_count
                data['replay_buffer_idx'] = self._tail
                data['use_count'] = 0
                data['collect_iter'] = data.get('collect_iter', -1)
                prio = data['priority'] if data['priority'] is not None else self._max_priority
            else:
                prio = self._max_priority
            weight = (max(prio, self._eps)) ** self.alpha
            self._sum_tree[self._tail] = weight
            self._min_tree[self._tail] = weight
            self._data[self._tail] = data
            self._tail = (self._tail + 1) % self._replay_buffer_size
            self._valid_count = min(self._valid_count + 1, self._replay_buffer_size)
            self._push_count += 1

    def sample(self, size: int, cur_learner_iter: int = -1, sample_range=None) -> Optional[List[Any]]:
        if size == 0:
            return []
        with self._lock:
            if self._valid_count < size:
                return None
            total = self._sum_tree.reduce(0, self._valid_count)
            mass = (np.random.rand(size) + np.arange(size)) / size * total
            indices = self._sum_tree.find_prefixsum_idx(mass)
            indices = np.clip(indices, 0, self._valid_count - 1)
            p_min = self._min_tree.reduce(0, self._valid_count) / total
            max_weight = (self._valid_count * p_min) ** (-self._beta)
            out = []
            for i in indices:
                d = self._data[int(i)]
                if isinstance(d, dict):
                    d = copy.copy(d)
                    p = self._sum_tree[int(i)] / total
                    d['IS'] = float(((self._valid_count * p) ** (-self._beta)) / max_weight)
                    d['use_count'] = d.get('use_count', 0) + 1
                    self._data[int(i)]['use_count'] = d['use_count']
                    if d['use_count'] >= self._max_use:
                        self._remove(int(i))
                out.append(d)
            if self._anneal_step != 0:
                self._beta = min(1.0, self._beta + self._beta_anneal_step * size)
            return out

    def _remove(self, idx: int) -> None:
        self._sum_tree[idx] = 0.0
        self._min_tree[idx] = float('inf')
        # keep slot content (simplest eviction: priority zero => never sampled)

    def update(self, info: dict) -> None:
        """info: {'replay_unique_id': [...], 'replay_buffer_idx': [...],
        'priority': [...]}"""
        with self._lock:
            if info is None or 'priority' in info and info['priority'] is None:
                return
            ids = info.get('replay_buffer_idx', [])
            prios = info.get('priority', [])
            for idx, p in zip(ids, prios):
                idx = int(idx)
                if self._data[idx] is None:
                    continue
                weight = (max(float(p), self._eps)) ** self.alpha
                self._sum_tree[idx] = weight
                self._min_tree[idx] = weight
                self._max_priority = max(self._max_priority, float(p))

    def clear(self) -> None:
        super().clear()
        capacity = int(np.power(2, np.ceil(np.log2(self._replay_buffer_size))))
        self._sum_tree = SumSegmentTree(capacity)
        self._min_tree = MinSegmentTree(capacity)
        self._max_priority = 1.0


@BUFFER_REGISTRY.register('episode')
class EpisodeReplayBuffer(NaiveReplayBuffer):
    """Stores whole episodes as items."""

    config = dict(
        type='episode',
        replay_buffer_size=10000,
        deepcopy=False,
        enable_track_used_data=False,
    )


def create_buffer(cfg: EasyDict, tb_logger=None, exp_name: str = 'default_experiment',
                  instance_name: str = 'buffer') -> NaiveReplayBuffer:
    cfg = EasyDict(cfg)
    buffer_type = cfg.get('type', 'naive')
    return BUFFER_REGISTRY.build(buffer_type, cfg=cfg, tb_logger=tb_logger, exp_name=exp_name,
                                 instance_name=instance_name)


def get_buffer_cls(cfg: EasyDict) -> type:
    return BUFFER_REGISTRY.get(cfg.get('type', 'naive'))


@BUFFER_REGISTRY.register('sequence')
class SequenceReplayBuffer(NaiveReplayBuffer):
    """FIFO buffer sampling CONSECUTIVE windows for sequence models
    (DreamerV3 world-model training).

    Parity: reference ding/worker/replay_buffer/naive_buffer.py
    SequenceNaiveReplayBuffer ('sequence':471): sample(batch, sequence, iter)
    -> list of ``batch`` lists, each ``sequence`` consecutive transitions.
    Windows never straddle the ring's write head (stale/fresh seam).
    """

    def sample(self, batch: int, sequence: int = 1, cur_learner_iter: int = -1,
               sample_range=None, replace: bool = False):
        if batch == 0:
            return []
        with self._lock:
            if self._valid_count < batch * sequence:
                return None
            n = self._valid_count
            # valid start positions: window fully inside [0, n) and, when the
            # ring has wrapped, not crossing the write head at self._tail
            starts = []
            tries = 0
            while len(starts) < batch and tries < 100 * batch:
                tries += 1
                s = np.random.randint(0, n - sequence + 1)
                if self._valid_count == self._replay_buffer_size:
                    # ring wrapped: logical order starts at tail
                    s = (self._tail + s) % n
                    end = s + sequence
                    if end > n:
                        continue  # physical wrap inside window: skip
                starts.append(s)
            return [[self._data[(s + j) % n] for j in range(sequence)] for s in starts]


@BUFFER_REGISTRY.register('elastic')
class ElasticReplayBuffer(NaiveReplayBuffer):
    """Naive buffer whose SAMPLEABLE window grows on a schedule:
    cfg.set_buffer_size(envstep) -> current window size (newest-first).

    Parity: reference ding/worker/replay_buffer/naive_buffer.py
    ElasticReplayBuffer ('elastic':381).
    """

    config = dict(
        type='elastic',
        replay_buffer_size=10000,
        deepcopy=False,
        enable_track_used_data=False,
        periodic_thruput_seconds=60,
        set_buffer_size=lambda envstep: 10000,
    )

    def __init__(self, cfg, tb_logger=None, exp_name: str = 'default_experiment', instance_name: str = 'buffer'):
        super().__init__(cfg, tb_logger, exp_name, instance_name)
        self._set_buffer_size = self._cfg.set_buffer_size
        self._current_buffer_size = self._set_buffer_size(0)

    def update_buffer_size(self, envstep: int) -> None:
        self._current_buffer_size = self._set_buffer_size(envstep)

    def sample(self, size: int, cur_learner_iter: int = -1, sample_range=None):
        if size == 0:
            return []
        with self._lock:
            window = min(self._valid_count, self._current_buffer_size)
            if window < size:
                return None
            # newest `window` entries of the ring
            idx = (self._tail - 1 - np.random.choice(window, size, replace=False)) % self._replay_buffer_size
            return [self._data[i] for i in idx]
