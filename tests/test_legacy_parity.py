"""Behavioral tests for the reference-parity surfaces added late in round 2:
legacy worker helpers, the extra nn blocks, the DequeBuffer legacy adapter
and the env conformance checkers.
"""
import numpy as np
import pytest
import torch

from ding.utils import EasyDict


def test_vector_eval_monitor_caps_per_env():
    from ding.worker import VectorEvalMonitor
    m = VectorEvalMonitor(3, 7)
    # env 0 tries to submit 10 fast episodes; cap must hold it to its share
    for _ in range(10):
        m.update_reward(0, 1.0)
    assert not m.is_finished()
    for eid in (1, 2):
        for _ in range(4):
            m.update_reward(eid, 2.0)
    assert m.is_finished()
    rets = m.get_episode_return()
    assert len(rets) == 7
    assert rets.count(1.0) <= 3, "fast env must not crowd out slow envs"


def test_to_tensor_transitions_aliases_next_obs():
    from ding.worker import to_tensor_transitions
    data = [{'obs': np.full(2, i, np.float32), 'next_obs': np.full(2, i + 1, np.float32),
             'reward': float(i), 'done': False} for i in range(4)]
    out = to_tensor_transitions(data)
    for i in range(3):
        assert out[i]['next_obs'] is out[i + 1]['obs'], "fragment must store each frame once"
    out2 = to_tensor_transitions(data, shallow_copy_next_obs=False)
    assert out2[0]['next_obs'] is not out2[1]['obs']


def test_sequence_replay_buffer_contiguous_windows():
    from ding.worker import SequenceReplayBuffer
    buf = SequenceReplayBuffer(EasyDict({'replay_buffer_size': 64}))
    buf.push(list(range(40)))
    seqs = buf.sample(6, sequence=5)
    assert len(seqs) == 6
    for s in seqs:
        assert [s[i + 1] - s[i] for i in range(4)] == [1] * 4, "windows must be consecutive"


def test_learner_hook_priority_order():
    from ding.worker import LearnerHook, add_learner_hook, merge_hooks

    class H(LearnerHook):

        def __call__(self, engine):
            pass

    hooks = {'after_iter': []}
    add_learner_hook(hooks, H('late', 90))
    add_learner_hook(hooks, H('early', 10))
    assert [h.name for h in hooks['after_iter']] == ['early', 'late']
    merged = merge_hooks(hooks, {'after_iter': [H('mid', 50)]})
    assert [h.name for h in merged['after_iter']] == ['early', 'mid', 'late']


def test_deque_buffer_wrapper_priority_update():
    from ding.data import DequeBufferWrapper
    cfg = EasyDict({**DequeBufferWrapper.config, 'priority': True, 'priority_IS_weight': True})
    w = DequeBufferWrapper(cfg)
    w.push([{'x': i, 'priority': 10.0 if i == 3 else 0.01} for i in range(16)])
    hits = sum(1 for _ in range(30) for d in w.sample(4) if d['x'] == 3)
    assert hits > 20, f"high-priority item under-sampled: {hits}"
    batch = w.sample(4)
    assert all('IS' in d for d in batch)
    w.update({'priority': [0.001] * 4})
    assert w.count() == 16


def test_extra_blocks_shapes_and_grads():
    from ding.torch_utils.network import (
        Swish, SoftArgmax, GumbelSoftmax, VectorMerge, GatingType, SumMerge, resnet18
    )
    x = torch.randn(3, 5, requires_grad=True)
    Swish()(x).sum().backward()
    assert x.grad is not None
    heat = torch.zeros(2, 1, 9, 9)
    heat[0, 0, 2, 7] = 50.0
    heat[1, 0, 8, 0] = 50.0
    coords = SoftArgmax()(heat)
    assert torch.allclose(coords[0], torch.tensor([2.0, 7.0]), atol=0.1)
    assert torch.allclose(coords[1], torch.tensor([8.0, 0.0]), atol=0.1)
    gs = GumbelSoftmax()(torch.randn(4, 6), hard=True)
    assert torch.allclose(gs.sum(-1), torch.ones(4))
    vm = VectorMerge({'a': 8, 'b': 4}, 16, GatingType.POINTWISE)
    out = vm({'a': torch.randn(5, 8), 'b': torch.randn(5, 4)})
    assert out.shape == (5, 16)
    assert SumMerge()([torch.ones(2, 2)] * 3).sum() == 12
    m = resnet18(num_classes=7, in_chans=1)
    assert m(torch.randn(2, 1, 48, 48)).shape == (2, 7)


def test_env_conformance_checkers_catch_violations():
    from ding.envs import check_all, check_different_memory
    from dizoo.classic_control.cartpole.envs.cartpole_env import CartPoleEnv
    env = CartPoleEnv(EasyDict({}))
    check_all(env)
    env.close()
    arr = np.zeros(3)
    with pytest.raises(AssertionError):
        check_different_memory(arr, arr, 1)  # aliased obs must be flagged


def test_policy_factory_random_forward():
    from ding.policy import PolicyFactory
    from ding.envs.common.spaces import Discrete

    class _Collect:

        def process_transition(self, *a):
            return {}

        def get_train_sample(self, x):
            return x

        def get_attribute(self, name):
            return None

    rp = PolicyFactory.get_random_policy(_Collect(), action_space=Discrete(5))
    out = rp.forward({0: np.zeros(4), 3: np.zeros(4)})
    assert set(out.keys()) == {0, 3}
    for v in out.values():
        assert 0 <= int(v['action']) < 5


def test_timestep_collate_nondestructive():
    """Re-collating the same stored samples must work: buffers hand out
    references, so collate must not pop 'prev_state' in place (bug found by
    the on-device QMIX learning check)."""
    from ding.utils.data.collate_fn import timestep_collate
    batch = [
        {
            'obs': [torch.randn(3) for _ in range(4)],
            'action': [torch.zeros(1, dtype=torch.long) for _ in range(4)],
            'prev_state': [None] * 4,
        } for _ in range(2)
    ]
    out1 = timestep_collate(batch)
    out2 = timestep_collate(batch)  # second pass over the SAME dicts
    assert 'prev_state' in batch[0], "collate must not mutate stored samples"
    assert out1['obs'].shape == out2['obs'].shape == (4, 2, 3)
