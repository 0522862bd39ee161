"""MBSAC / STEVE-SAC on a toy differentiable world model."""
import pytest
import torch

from ding.utils import EasyDict, deep_merge_dicts


class ToyDreamModel(torch.nn.Module):
    """Linear dynamics dream model: obs' = A obs + B a, r = -|obs|^2 row-sum."""

    def __init__(self, obs_dim=3, act_dim=1, horizon=3):
        super().__init__()
        self.A = torch.nn.Parameter(torch.eye(obs_dim) * 0.9, requires_grad=True)
        self.B = torch.nn.Parameter(torch.randn(act_dim, obs_dim) * 0.1, requires_grad=True)
        self._horizon = horizon

    def rollout_length_scheduler(self, envstep):
        return self._horizon

    def step(self, obs, action, **kwargs):
        nxt = obs @ self.A + action @ self.B.to(action.dtype)
        reward = -(nxt ** 2).sum(-1)
        done = torch.zeros_like(reward)
        return reward, nxt, done

    # DreamWorldModel.rollout, reused verbatim
    from ding.world_model.base_world_model import DreamWorldModel
    rollout = DreamWorldModel.rollout


def _sac_cfg(policy_cls, extra=None):
    cfg = EasyDict(deep_merge_dicts(policy_cls.default_config(), EasyDict(dict(
        cuda=False,
        model=dict(obs_shape=3, action_shape=1, twin_critic=True, action_space='reparameterization'),
        learn=dict(update_per_collect=1, batch_size=8, auto_alpha=False, alpha=0.2),
        collect=dict(n_sample=8, unroll_len=1),
        other=dict(replay_buffer=dict(replay_buffer_size=100)),
    ))))
    if extra:
        cfg = EasyDict(deep_merge_dicts(cfg, EasyDict(extra)))
    return cfg


def _fake_transitions(n=8, obs_dim=3, act_dim=1):
    return [
        {
            'obs': torch.randn(obs_dim),
            'next_obs': torch.randn(obs_dim),
            'action': torch.randn(act_dim),
            'reward': torch.randn(1),
            'done': False,
            'collect_iter': 0,
        } for _ in range(n)
    ]


def test_mbsac_learn_step():
    from ding.policy import MBSACPolicy
    pol = MBSACPolicy(_sac_cfg(MBSACPolicy), enable_field=['learn'])
    wm = ToyDreamModel()
    before = [p.clone() for p in pol._model.actor.parameters()]
    info = pol._forward_learn(_fake_transitions(), world_model=wm, envstep=10)
    assert all(torch.isfinite(torch.tensor(v)) for k, v in info.items() if isinstance(v, float))
    changed = any(not torch.allclose(a, b) for a, b in zip(before, pol._model.actor.parameters()))
    assert changed, "actor did not update"


def test_mbsac_sample_transition_mode():
    from ding.policy import MBSACPolicy
    pol = MBSACPolicy(_sac_cfg(MBSACPolicy, dict(learn=dict(sample_state=False))), enable_field=['learn'])
    wm = ToyDreamModel()
    info = pol._forward_learn(_fake_transitions(), world_model=wm, envstep=10)
    assert 'critic_loss' in info and 'policy_loss' in info


def test_stevesac_learn_step():
    from ding.policy import STEVESACPolicy
    pol = STEVESACPolicy(_sac_cfg(STEVESACPolicy, dict(learn=dict(ensemble_size=2))), enable_field=['learn'])
    wm = ToyDreamModel()
    info = pol._forward_learn(_fake_transitions(), world_model=wm, envstep=10)
    assert 'critic_loss' in info and 'target_q_value' in info


def test_dream_rollout_contract():
    """rollout returns [N+1,B,O] obss, [N+1,B,A] actions, [N,B] rewards,
    [N+1,B] aug, [N,B] dones and keeps grads to the policy only."""
    wm = ToyDreamModel(horizon=4)
    lin = torch.nn.Linear(3, 1)

    def actor_fn(obs):
        a = torch.tanh(lin(obs))
        return a, torch.zeros(obs.shape[0])

    obs = torch.randn(5, 3)
    obss, actions, rewards, aug, dones = wm.rollout(obs, actor_fn, envstep=0)
    assert obss.shape == (5, 5, 3) and actions.shape == (5, 5, 1)
    assert rewards.shape == (4, 5) and aug.shape == (5, 5) and dones.shape == (4, 5)
    rewards.sum().backward()
    assert lin.weight.grad is not None
    # world model was re-enabled for training after rollout
    assert all(p.requires_grad for p in wm.parameters())
    # but received no grads from the policy rollout
    assert wm.A.grad is None
