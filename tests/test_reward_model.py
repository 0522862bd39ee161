"""Reward-model subsystem tests: train + estimate round-trips."""
import numpy as np
import pytest
import torch

from ding.reward_model import (
    create_reward_model, RndRewardModel, ICMRewardModel, GailRewardModel, GuidedCostRewardModel, PwilRewardModel,
    RedRewardModel, PdeilRewardModel, TrexRewardModel, HerRewardModel, RndNGURewardModel, EpisodicNGURewardModel,
)
from ding.utils import EasyDict


def make_transitions(n=32, obs_dim=4, discrete=True):
    out = []
    for _ in range(n):
        out.append({
            'obs': torch.randn(obs_dim),
            'next_obs': torch.randn(obs_dim),
            'action': torch.tensor(np.random.randint(0, 2)) if discrete else torch.randn(1),
            'reward': torch.tensor([1.0]),
            'done': False,
        })
    return out


def test_rnd():
    rm = RndRewardModel(EasyDict(dict(obs_shape=4, hidden_size_list=[16, 16], update_per_collect=2)))
    data = make_transitions()
    rm.collect_data(data)
    rm.train()
    est = rm.estimate(data)
    assert len(est) == len(data)
    assert not torch.allclose(est[0]['reward'], data[0]['reward'])  # modified copy
    assert torch.allclose(data[0]['reward'], torch.tensor([1.0]))  # original untouched
    rm.clear_data()
    assert rm.train_obs == []


def test_icm():
    rm = ICMRewardModel(EasyDict(dict(obs_shape=4, action_shape=2, hidden_size_list=[16, 16], update_per_collect=2)))
    data = make_transitions()
    rm.collect_data(data)
    rm.train()
    est = rm.estimate(data)
    assert len(est) == len(data)


def test_gail_and_gcl():
    data = make_transitions(discrete=False)
    expert = make_transitions(discrete=False)
    gail = GailRewardModel(EasyDict(dict(input_size=5, update_per_collect=2)))
    gail.load_expert_data(expert)
    gail.collect_data(data)
    gail.train()
    est = gail.estimate(data)
    assert (torch.stack([e['reward'] for e in est]) >= 0).all()
    gcl = GuidedCostRewardModel(EasyDict(dict(input_size=5, update_per_collect=2)))
    gcl.load_expert_data(expert)
    gcl.collect_data(data)
    gcl.train()
    assert len(gcl.estimate(data)) == len(data)


def test_nonparametric_models():
    data = make_transitions(discrete=False)
    expert = make_transitions(discrete=False)
    for cls, cfg in ((PwilRewardModel, {}), (RedRewardModel, dict(input_size=5, update_per_collect=2)),
                     (PdeilRewardModel, {})):
        rm = cls(EasyDict(cfg))
        rm.load_expert_data(expert)
        rm.train()
        assert len(rm.estimate(data)) == len(data)


def test_trex():
    rm = TrexRewardModel(EasyDict(dict(input_size=4, update_per_collect=4, min_snippet_length=3,
                                       max_snippet_length=6)))
    trajs = [make_transitions(10) for _ in range(4)]
    rm.load_ranked_trajectories(trajs)
    rm.train()
    assert len(rm.estimate(make_transitions(8))) == 8


def test_her():
    rm = HerRewardModel(dict(her_strategy='future', her_replay_k=2))
    episode = make_transitions(6)
    relabelled = rm.estimate(episode)
    assert len(relabelled) == 2 and len(relabelled[0]) == 6


def test_ngu():
    rnd = RndNGURewardModel(EasyDict(dict(obs_shape=4, hidden_size_list=[16, 16], update_per_collect=2)))
    epi = EpisodicNGURewardModel(EasyDict(dict(obs_shape=4, action_shape=2, hidden_size_list=[16, 16],
                                               update_per_collect=2)))
    data = make_transitions()
    rnd.collect_data(data)
    rnd.train()
    alpha = rnd.estimate(data)
    epi.collect_data(data)
    epi.train()
    er = epi.estimate(data)
    from ding.reward_model.ngu_reward_model import fusion_reward
    fused = fusion_reward(data, er, alpha)
    assert len(fused) == len(data)


def test_factory():
    rm = create_reward_model(EasyDict(dict(type='rnd', obs_shape=4, hidden_size_list=[16, 16])))
    assert isinstance(rm, RndRewardModel)


def test_her_bitflip_relabel():
    """HER on BitFlip: relabeling with 'final' strategy turns a failed
    episode's last transition into a success (reward 1)."""
    import torch
    from ding.reward_model import HerRewardModel
    from dizoo.bitflip.envs import BitFlipEnv

    env = BitFlipEnv({'n_bits': 6})
    env.seed(3, dynamic_seed=False)
    obs = env.reset()
    episode = []
    rng = __import__('numpy').random.RandomState(0)
    for _ in range(12):
        action = int(rng.randint(0, 6))
        ts = env.step(action)
        episode.append({
            'obs': torch.as_tensor(obs), 'next_obs': torch.as_tensor(ts.obs),
            'action': torch.tensor([action]), 'reward': torch.as_tensor(ts.reward), 'done': ts.done,
        })
        obs = ts.obs
        if ts.done:
            break

    def goal_fn(t):
        # achieved goal = the state bits of next_obs
        return t['next_obs'][:6]

    def reward_fn(goal, t):
        ok = torch.allclose(t['next_obs'][:6].float(), torch.as_tensor(goal).float())
        return torch.ones_like(t['reward']) if ok else torch.zeros_like(t['reward'])

    her = HerRewardModel({'her_strategy': 'final', 'her_replay_k': 2,
                          'goal_fn': goal_fn, 'reward_fn': reward_fn})
    relabelled = her.estimate(episode)
    assert len(relabelled) == 2
    # final-strategy: last transition always achieves the (relabelled) goal
    for ep in relabelled:
        assert float(ep[-1]['reward']) == 1.0
