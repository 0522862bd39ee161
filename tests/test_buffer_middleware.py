"""Behavioral matrix for the Buffer middleware chain (reference
ding/data/buffer/tests): PER sampling/update, use-count eviction,
staleness eviction, sample-range view, clone_object isolation,
group sampling, padding.
"""
import copy

import pytest
import torch

from ding.data.buffer import DequeBuffer
from ding.data.buffer.middleware import (
    PriorityExperienceReplay, clone_object, use_time_check, staleness_check, sample_range_view, padding,
    group_sample,
)


def _push_n(buf, n=20, **meta_fn):
    for i in range(n):
        meta = {k: fn(i) for k, fn in meta_fn.items()}
        buf.push({'obs': torch.tensor([float(i)]), 'i': i}, meta=meta)


def test_per_priority_bias_and_update():
    buf = DequeBuffer(size=64)
    per = PriorityExperienceReplay(buf, IS_weight=True)
    buf.use(per)
    for i in range(32):
        buf.push({'i': i}, meta={'priority': 10.0 if i == 7 else 0.01})
    counts = 0
    for _ in range(50):
        batch = buf.sample(4, replace=True)
        counts += sum(1 for b in batch if b.data['i'] == 7)
        for b in batch:
            assert 'priority_IS' in b.meta
    assert counts > 30, f"high-priority item under-sampled: {counts}"
    # lowering its priority must stop the bias
    for b in buf.sample(32, replace=True):
        if b.data['i'] == 7:
            buf.update(b.index, b.data, {'priority': 0.001})
    counts2 = sum(
        1 for _ in range(30) for b in buf.sample(4, replace=True) if b.data['i'] == 7
    )
    assert counts2 < counts


def test_use_time_check_evicts():
    buf = DequeBuffer(size=64)
    buf.use(use_time_check(buf, max_use=2))
    _push_n(buf, 4)
    assert buf.count() == 4
    for _ in range(2):
        buf.sample(4)
    # every item used twice -> evicted
    assert buf.count() == 0


def test_staleness_check_evicts_on_sample():
    buf = DequeBuffer(size=64)
    buf.use(staleness_check(buf, max_staleness=5))
    for i in range(6):
        buf.push({'i': i}, meta={'train_iter_data_collected': 0})
    out = buf.sample(3, train_iter_sample_data=3)
    assert len(out) == 3
    with pytest.raises(ValueError):
        # staleness 10 > 5: all evicted, sample must fail
        buf.sample(3, train_iter_sample_data=10)
    assert buf.count() == 0


def test_sample_range_view_restricts():
    buf = DequeBuffer(size=64)
    buf.use(sample_range_view(buf, start=-5, end=None))
    _push_n(buf, 20)
    for _ in range(10):
        batch = buf.sample(3)
        assert all(b.data['i'] >= 15 for b in batch)


def test_clone_object_isolation():
    buf = DequeBuffer(size=8)
    buf.use(clone_object())
    src = {'x': torch.zeros(2)}
    buf.push(src)
    src['x'] += 99  # mutate after push
    got = buf.sample(1)[0].data
    assert torch.all(got['x'] == 0), "clone_object must deep-copy on push"
    got['x'] += 7  # mutate the sample
    again = buf.sample(1)[0].data
    assert torch.all(again['x'] == 0), "clone_object must deep-copy on sample"


def test_group_sample():
    buf = DequeBuffer(size=64)
    buf.use(group_sample(size_in_group=4, ordered_in_group=True, max_use_in_group=True))
    for env in range(3):
        for t in range(8):
            buf.push({'env': env, 't': t}, meta={'env_episode': env})
    groups = buf.sample(2, groupby='env_episode')
    assert len(groups) == 2
    for g in groups:
        assert len(g) == 4
        ts = [b.data['t'] for b in g]
        assert ts == sorted(ts), "ordered_in_group must keep time order"
        assert len({b.data['env'] for b in g}) == 1


def test_padding_equalizes_groups():
    buf = DequeBuffer(size=64)
    buf.use(padding())
    for env, n in ((0, 3), (1, 5)):
        for t in range(n):
            buf.push({'env': env, 't': t}, meta={'env_episode': env})
    groups = buf.sample(2, groupby='env_episode')
    sizes = {len(g) for g in groups}
    assert sizes == {5}, f"padding must equalize group sizes, got {sizes}"
