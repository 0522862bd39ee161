"""IBC/BCQ/TD3-VAE/prompt/NGU policy construction + one learn step."""
import pytest
import torch

from ding.utils import EasyDict, deep_merge_dicts


def _mk(policy_cls, extra):
    cfg = policy_cls.default_config()
    cfg = EasyDict(deep_merge_dicts(cfg, extra))
    return policy_cls(cfg, enable_field=['learn'])


def _transitions(n=16, obs=3, act=1, discrete=False):
    out = []
    for _ in range(n):
        out.append({
            'obs': torch.randn(obs),
            'next_obs': torch.randn(obs),
            'action': torch.randint(0, 2, ()).long() if discrete else torch.tanh(torch.randn(act)),
            'reward': torch.tensor([0.5]),
            'done': False,
        })
    return out


def test_ibc():
    from ding.policy import IBCPolicy
    policy = _mk(IBCPolicy, dict(model=dict(obs_shape=3, action_shape=1, hidden_size=32, hidden_layer_num=2)))
    out = policy._forward_learn(_transitions())
    assert 'total_loss' in out
    policy._init_eval()
    acts = policy._forward_eval({0: torch.randn(3), 1: torch.randn(3)})
    assert acts[0]['action'].shape == (1, )


def test_bcq():
    from ding.policy import BCQPolicy
    policy = _mk(BCQPolicy, dict(model=dict(obs_shape=3, action_shape=1, action_space='regression',
                                            twin_critic=True)))
    out = policy._forward_learn(_transitions())
    assert 'vae_loss' in out
    policy._init_eval()
    acts = policy._forward_eval({0: torch.randn(3)})
    assert acts[0]['action'].shape == (1, )


def test_td3_vae():
    from ding.policy import TD3VAEPolicy
    policy = _mk(TD3VAEPolicy, dict(original_action_shape=1,
                                    model=dict(obs_shape=3, action_shape=2, action_space='regression',
                                               twin_critic=True)))
    out = policy.train_vae(_transitions())
    assert 'vae_loss' in out


def test_prompt_pg():
    from ding.policy import PromptPGPolicy
    policy = _mk(PromptPGPolicy, dict(model=dict(embedding_size=32)))
    obs = {'train_sample': 'what is two plus two', 'candidate_samples': ['example a', 'example b', 'example c']}
    data = [{'obs': obs, 'action': torch.tensor([1]), 'return': 1.0}]
    out = policy._forward_learn(data)
    assert 'total_loss' in out
    policy._init_eval()
    res = policy._forward_eval({0: obs})
    assert 'action' in res[0]


def test_ngu_r2d3_registered():
    from ding.utils import POLICY_REGISTRY
    for name in ('ngu', 'r2d3', 'r2d2_gtrxl', 'ibc', 'bcq', 'td3_vae', 'prompt_pg', 'prompt_awr', 'pc_bfs'):
        assert name in POLICY_REGISTRY
