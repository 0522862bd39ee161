import numpy as np
import pytest
import torch

from ding.utils import (
    EasyDict, SumSegmentTree, MinSegmentTree, deep_merge_dicts, lists_to_dicts, dicts_to_lists,
    squeeze, set_pkg_seed, split_data_generator, Registry, RunningMeanStd, EasyTimer,
    get_data_compressor, get_data_decompressor,
)


def test_easydict():
    d = EasyDict({"a": 1, "b": {"c": [2, {"d": 3}]}})
    assert d.a == 1 and d.b.c[1].d == 3
    d.x = {"y": 5}
    assert d.x.y == 5
    import copy
    d2 = copy.deepcopy(d)
    d2.b.c[1].d = 9
    assert d.b.c[1].d == 3


def test_registry():
    R = Registry("test")

    @R.register("foo")
    class Foo:
        pass

    assert R.get("foo") is Foo
    assert "foo" in R
    with pytest.raises(KeyError):
        @R.register("foo")
        class Bar:
            pass


def test_segment_tree_vs_numpy():
    rng = np.random.RandomState(0)
    cap = 128
    st = SumSegmentTree(cap)
    vals = rng.rand(cap)
    st[np.arange(cap)] = vals
    assert abs(st.reduce() - vals.sum()) < 1e-9
    assert abs(st.reduce(10, 50) - vals[10:50].sum()) < 1e-9
    # prefix-sum sampling correctness
    for p in [0.0, vals.sum() * 0.3, vals.sum() * 0.999]:
        i = st.find_prefixsum_idx(p)
        cs = np.cumsum(vals)
        expect = int(np.searchsorted(cs, p, side="right"))
        assert i == expect
    mt = MinSegmentTree(cap)
    mt[np.arange(cap)] = vals
    assert abs(mt.reduce(5, 77) - vals[5:77].min()) < 1e-12


def test_dict_list_helpers():
    data = [{"a": 1, "b": 2}, {"a": 3, "b": 4}]
    d = lists_to_dicts(data)
    assert d == {"a": [1, 3], "b": [2, 4]}
    assert dicts_to_lists(d) == data
    assert squeeze((4, )) == 4
    merged = deep_merge_dicts({"a": {"b": 1, "c": 2}}, {"a": {"b": 7}})
    assert merged == {"a": {"b": 7, "c": 2}}


def test_split_data_generator():
    data = {"obs": torch.arange(10).float(), "scalar": 3}
    batches = list(split_data_generator(data, 5, shuffle=False))
    assert len(batches) == 2 and batches[0]["obs"].shape[0] == 5
    assert batches[0]["scalar"] == 3


def test_running_mean_std():
    rms = RunningMeanStd(shape=(3, ))
    data = np.random.RandomState(1).randn(1000, 3) * 2 + 5
    for i in range(0, 1000, 100):
        rms.update(data[i:i + 100])
    assert np.allclose(rms.mean, data.mean(0), atol=1e-2)
    assert np.allclose(rms.std, data.std(0), atol=1e-2)


def test_compression_roundtrip():
    payload = {"x": np.arange(100), "y": [1, "two"]}
    for name in ("none", "zlib", "lz4"):
        c = get_data_compressor(name)(payload)
        out = get_data_decompressor(name)(c)
        assert out["y"] == payload["y"] and (out["x"] == payload["x"]).all()


def test_timer():
    with EasyTimer(cuda=False) as t:
        sum(range(1000))
    assert t.value >= 0


def test_seed():
    set_pkg_seed(42, use_cuda=False)
    a = torch.rand(3)
    set_pkg_seed(42, use_cuda=False)
    assert torch.equal(a, torch.rand(3))


def test_env_supervisor():
    from ding.envs import EnvSupervisor
    from ding.framework.supervisor import ChildType
    from dizoo.classic_control.cartpole.envs.cartpole_env import CartPoleEnv
    sup = EnvSupervisor(ChildType.THREAD, env_fn=[lambda: CartPoleEnv({}) for _ in range(2)])
    sup.seed(0)
    sup.launch()
    obs = sup.ready_obs
    assert len(obs) == 2
    import numpy as np
    for _ in range(5):
        ts = sup.step({i: np.random.randint(2) for i in sup.ready_obs_id})
    sup.close()


def test_new_env_wrappers():
    """StaticObsNorm / Ram / Transpose / ObsPlusPrevActRew / AllinObs /
    GymToGymnasium wrapper semantics."""
    import numpy as np
    from ding.envs.env_wrappers.env_wrappers import (
        AllinObsWrapper, GymToGymnasiumWrapper, ObsPlusPrevActRewWrapper, RamWrapper, StaticObsNormWrapper,
        TransposeWrapper,
    )

    class FakeEnv:

        def __init__(self, obs):
            self._obs = np.asarray(obs, dtype=np.float32)

        def reset(self, **kw):
            return self._obs

        def step(self, action):
            return self._obs, 1.0, False, {}

    e = StaticObsNormWrapper(FakeEnv([10.0, 20.0]), mean=[10.0, 20.0], std=[2.0, 4.0])
    assert np.allclose(e.reset(), [0, 0])
    obs, r, d, i = e.step(0)
    assert np.allclose(obs, [0, 0]) and r == 1.0

    e = RamWrapper(FakeEnv(np.arange(8)))
    assert e.reset().shape == (8, 1, 1)

    e = TransposeWrapper(FakeEnv(np.zeros((4, 5, 3))))
    assert e.reset().shape == (3, 4, 5)

    e = ObsPlusPrevActRewWrapper(FakeEnv([1.0]))
    o = e.reset()
    assert o['prev_action'] == -1 and o['prev_reward_extrinsic'] == 0.0
    o, _, _, _ = e.step(2)
    o2, _, _, _ = e.step(3)
    assert o2['prev_action'] == 2 and o2['prev_reward_extrinsic'] == 1.0

    e = AllinObsWrapper(FakeEnv([1.0]))
    o = e.reset()
    assert set(o.keys()) == {'obs', 'reward'}

    class GymnasiumEnv:

        def reset(self, seed=None, **kw):
            return np.zeros(2), {}

        def step(self, action):
            return np.zeros(2), 0.5, False, True, {}

    e = GymToGymnasiumWrapper(GymnasiumEnv())
    e.seed(3)
    assert e.reset().shape == (2, )
    obs, r, done, info = e.step(0)
    assert done is True and r == 0.5  # truncated folds into done
