"""Model-based SAC family: MBSAC (model-based value expansion + value
gradients over TD-lambda returns) and STEVE-SAC (stochastic ensemble value
expansion with inverse-variance weighting).

Parity: reference ding/policy/mbpolicy/mbsac.py ('mbsac':18, 'stevesac':220)
and mbpolicy/utils.py (q_evaluation).
"""
from functools import partial
from typing import Any, Dict, List

import torch
import torch.nn as nn
from torch.distributions import Independent, Normal

from ding.rl_utils import generalized_lambda_returns
from ding.torch_utils import fold_batch, to_device, unfold_batch, unsqueeze_repeat
from ding.utils import POLICY_REGISTRY
from .common_utils import default_preprocess_learn
from .sac import SACPolicy


def q_evaluation(obss: torch.Tensor, actions: torch.Tensor, q_critic_fn) -> Any:
    """Evaluate Q(s_t, a_t) along a [N, B, ...] trajectory by folding time
    into the batch dim (ONE batched critic pass on the GPU instead of N)."""
    obss, dim = fold_batch(obss, 1)
    actions, _ = fold_batch(actions, 1)
    q_values = q_critic_fn(obss, actions)
    if isinstance(q_values, (list, tuple)):
        return [unfold_batch(q, dim) for q in q_values]
    return unfold_batch(q_values, dim)


def _tanh_gaussian_actor_fn(learn_model, alpha):
    """(action, -alpha*logp) with tanh-squash log-prob correction."""

    def actor_fn(obs: torch.Tensor):
        (mu, sigma) = learn_model.forward(obs, mode='compute_actor')['logit']
        dist = Independent(Normal(mu, sigma), 1)
        pred = dist.rsample()
        action = torch.tanh(pred)
        log_prob = dist.log_prob(pred) + 2 * (
            pred + torch.nn.functional.softplus(-2. * pred) - torch.log(torch.tensor(2.))
        ).sum(-1)
        return action, -alpha.detach() * log_prob

    return actor_fn


@POLICY_REGISTRY.register('mbsac')
class MBSACPolicy(SACPolicy):
    """SAC trained on differentiable world-model rollouts: critic regresses
    TD-lambda targets over the imagined trajectory (value expansion), actor
    maximizes the lambda-return through the dynamics (value gradients)."""

    config = dict(
        learn=dict(
            lambda_=0.8,
            grad_clip=100,
            sample_state=True,
        ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        self._target_model.requires_grad_(False)
        self._lambda = self._cfg.learn.lambda_
        self._grad_clip = self._cfg.learn.grad_clip
        self._sample_state = self._cfg.learn.sample_state
        assert not self._auto_alpha, "mbsac: auto_alpha not supported"
        self._actor_fn = _tanh_gaussian_actor_fn(self._learn_model, self._alpha)

        def critic_fn(obss, actions, model):
            q = model.forward({'obs': obss, 'action': actions}, mode='compute_critic')['q_value']
            return q

        self._critic_fn = critic_fn
        self._forward_learn_cnt = 0

    def _forward_learn(self, data: dict, world_model=None, envstep: int = 0) -> Dict[str, Any]:
        assert world_model is not None, "mbsac expects learner.train(data, envstep, policy_kwargs={'world_model':...})"
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight,
            ignore_done=self._cfg.learn.ignore_done, use_nstep=False
        )
        if self._cuda:
            data = to_device(data, self._device)
        if len(data['action'].shape) == 1:
            data['action'] = data['action'].unsqueeze(1)
        self._learn_model.train()
        self._target_model.train()

        if self._sample_state:
            obss, actions, rewards, aug_rewards, dones = \
                world_model.rollout(data['obs'], self._actor_fn, envstep)
        else:
            obss, actions, rewards, aug_rewards, dones = \
                world_model.rollout(data['next_obs'], self._actor_fn, envstep)
            obss = torch.cat([data['obs'].unsqueeze(0), obss])
            actions = torch.cat([data['action'].unsqueeze(0), actions])
            rewards = torch.cat([data['reward'].unsqueeze(0), rewards])
            aug_rewards = torch.cat([torch.zeros_like(data['reward']).unsqueeze(0), aug_rewards])
            dones = torch.cat([data['done'].unsqueeze(0), dones])
        dones = torch.cat([torch.zeros_like(dones[0]).unsqueeze(0), dones]).float()

        # (T+1, B) target values + entropy bonus
        target_q_values = q_evaluation(obss, actions, partial(self._critic_fn, model=self._target_model))
        if self._twin_critic:
            target_q_values = torch.min(target_q_values[0], target_q_values[1]) + aug_rewards
        else:
            target_q_values = target_q_values + aug_rewards
        # (T, B)
        lambda_return = generalized_lambda_returns(target_q_values, rewards, self._gamma, self._lambda, dones[1:])
        # mask imagined steps after termination
        weight = (1 - dones[:-1].detach()).cumprod(dim=0)

        q_values = q_evaluation(obss.detach(), actions.detach(), partial(self._critic_fn, model=self._learn_model))
        if self._twin_critic:
            critic_loss = 0.5 * torch.square(q_values[0][:-1] - lambda_return.detach()) \
                + 0.5 * torch.square(q_values[1][:-1] - lambda_return.detach())
        else:
            critic_loss = 0.5 * torch.square(q_values[:-1] - lambda_return.detach())
        critic_loss = (critic_loss * weight).mean()
        policy_loss = -(lambda_return * weight).mean()

        norm_dict = self._update({'critic_loss': critic_loss, 'policy_loss': policy_loss})
        self._forward_learn_cnt += 1
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr_q': self._optimizer_q.defaults['lr'],
            'cur_lr_p': self._optimizer_policy.defaults['lr'],
            'alpha': self._alpha.item(),
            'target_q_value': target_q_values.detach().mean().item(),
            'critic_loss': critic_loss.item(),
            'policy_loss': policy_loss.item(),
            **norm_dict,
        }

    def _update(self, loss_dict) -> Dict[str, float]:
        self._optimizer_q.zero_grad()
        loss_dict['critic_loss'].backward(retain_graph=True)
        critic_norm = nn.utils.clip_grad_norm_(self._model.critic.parameters(), self._grad_clip)
        self._optimizer_q.step()
        self._optimizer_policy.zero_grad()
        loss_dict['policy_loss'].backward()
        policy_norm = nn.utils.clip_grad_norm_(self._model.actor.parameters(), self._grad_clip)
        self._optimizer_policy.step()
        return {'policy_norm': float(policy_norm), 'critic_norm': float(critic_norm)}

    def _monitor_vars_learn(self) -> List[str]:
        return [
            'policy_loss', 'critic_loss', 'policy_norm', 'critic_norm', 'cur_lr_q', 'cur_lr_p', 'alpha',
            'target_q_value'
        ]


@POLICY_REGISTRY.register('stevesac')
class STEVESACPolicy(SACPolicy):
    """STEVE: rollouts through an ENSEMBLE of world models; per-horizon
    returns are weighted by inverse ensemble variance so unreliable long
    rollouts contribute less."""

    config = dict(
        learn=dict(
            grad_clip=100,
            ensemble_size=1,
        ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        self._target_model.requires_grad_(False)
        self._grad_clip = self._cfg.learn.grad_clip
        self._ensemble_size = self._cfg.learn.ensemble_size
        assert not self._auto_alpha, "stevesac: auto_alpha not supported"

        inner = _tanh_gaussian_actor_fn(self._learn_model, self._alpha)

        def actor_fn(obs: torch.Tensor):
            obs, dim = fold_batch(obs, 1)
            action, aug_reward = inner(obs)
            return unfold_batch(action, dim), unfold_batch(aug_reward, dim)

        self._actor_fn = actor_fn

        def critic_fn(obss, actions, model):
            return model.forward({'obs': obss, 'action': actions}, mode='compute_critic')['q_value']

        self._critic_fn = critic_fn
        self._forward_learn_cnt = 0

    def _forward_learn(self, data: dict, world_model=None, envstep: int = 0) -> Dict[str, Any]:
        assert world_model is not None
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight,
            ignore_done=self._cfg.learn.ignore_done, use_nstep=False
        )
        if self._cuda:
            data = to_device(data, self._device)
        if len(data['action'].shape) == 1:
            data['action'] = data['action'].unsqueeze(1)
        # [B, ...] -> [E, B, ...]
        data['next_obs'] = unsqueeze_repeat(data['next_obs'], self._ensemble_size)
        data['reward'] = unsqueeze_repeat(data['reward'], self._ensemble_size)
        data['done'] = unsqueeze_repeat(data['done'], self._ensemble_size)
        self._learn_model.train()
        self._target_model.train()

        obss, actions, rewards, aug_rewards, dones = \
            world_model.rollout(data['next_obs'], self._actor_fn, envstep, keep_ensemble=True)
        rewards = torch.cat([data['reward'].unsqueeze(0), rewards])
        dones = torch.cat([data['done'].unsqueeze(0), dones])

        # (T+1, E, B) — q_evaluation folds all leading dims, so the ensemble
        # axis rides along in the same single batched critic pass
        target_q_values = q_evaluation(obss, actions, partial(self._critic_fn, model=self._target_model))
        if self._twin_critic:
            target_q_values = torch.min(target_q_values[0], target_q_values[1]) + aug_rewards
        else:
            target_q_values = target_q_values + aug_rewards
        # inverse-variance-weighted STEVE return
        discounts = ((1 - dones) * self._gamma).cumprod(dim=0)
        discounts = torch.cat([torch.ones_like(discounts)[:1], discounts])
        cum_rewards = (rewards * discounts[:-1]).cumsum(dim=0)
        discounted_q_values = target_q_values * discounts[1:]
        steve_return = cum_rewards + discounted_q_values  # (T, E, B)
        steve_return_mean = steve_return.mean(1)
        with torch.no_grad():
            steve_return_inv_var = 1 / (1e-8 + steve_return.var(1, unbiased=False))
            steve_return_weight = steve_return_inv_var / (1e-8 + steve_return_inv_var.sum(dim=0))
        steve_return = (steve_return_mean * steve_return_weight).sum(0)  # (B,)

        q_values = self._learn_model.forward(
            {'obs': data['obs'], 'action': data['action']}, mode='compute_critic'
        )['q_value']
        if self._twin_critic:
            critic_loss = 0.5 * torch.square(q_values[0] - steve_return.detach()) \
                + 0.5 * torch.square(q_values[1] - steve_return.detach())
        else:
            critic_loss = 0.5 * torch.square(q_values - steve_return.detach())
        critic_loss = critic_loss.mean()
        policy_loss = -steve_return.mean()

        norm_dict = self._update({'critic_loss': critic_loss, 'policy_loss': policy_loss})
        self._forward_learn_cnt += 1
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr_q': self._optimizer_q.defaults['lr'],
            'cur_lr_p': self._optimizer_policy.defaults['lr'],
            'alpha': self._alpha.item(),
            'target_q_value': target_q_values.detach().mean().item(),
            'critic_loss': critic_loss.item(),
            'policy_loss': policy_loss.item(),
            **norm_dict,
        }

    _update = MBSACPolicy._update

    def _monitor_vars_learn(self) -> List[str]:
        return [
            'policy_loss', 'critic_loss', 'policy_norm', 'critic_norm', 'cur_lr_q', 'cur_lr_p', 'alpha',
            'target_q_value'
        ]
