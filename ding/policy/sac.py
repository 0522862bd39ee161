"""SAC policies: continuous SAC, discrete SAC, SQIL-SAC.

Parity: reference ding/policy/sac.py ('sac', 'discrete_sac', 'sqil_sac',
1,491 LoC).
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import numpy as np
import torch
import torch.nn.functional as F
from torch.distributions import Independent, Normal

from ding.model import model_wrap
from ding.rl_utils import get_train_sample, q_v_1step_td_data, q_v_1step_td_error
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn


@POLICY_REGISTRY.register('sac')
class SACPolicy(Policy):

    config = dict(
        type='sac',
        cuda=False,
        on_policy=False,
        multi_agent=False,
        priority=False,
        priority_IS_weight=False,
        random_collect_size=10000,
        transition_with_policy_data=True,
        model=dict(twin_critic=True, action_space='reparameterization'),
        learn=dict(
            update_per_collect=1,
            batch_size=256,
            learning_rate_q=3e-4,
            learning_rate_policy=3e-4,
            learning_rate_alpha=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            alpha=0.2,
            auto_alpha=True,
            log_space=True,
            ignore_done=False,
            target_entropy=None,
        ),
        collect=dict(unroll_len=1, ),
        eval=dict(),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000, ), ),
    )

    def default_model(self) -> tuple:
        return 'continuous_qac', ['ding.model.template.qac']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._twin_critic = self._cfg.model.twin_critic
        self._optimizer_q = Adam(self._model.critic.parameters(), lr=self._cfg.learn.learning_rate_q)
        self._optimizer_policy = Adam(self._model.actor.parameters(), lr=self._cfg.learn.learning_rate_policy)
        self._gamma = self._cfg.learn.discount_factor
        # entropy temperature
        if self._cfg.learn.auto_alpha:
            if self._cfg.learn.target_entropy is None:
                action_shape = self._cfg.model.action_shape
                self._target_entropy = -float(np.prod(action_shape if not np.isscalar(action_shape) else [action_shape]))
            else:
                self._target_entropy = self._cfg.learn.target_entropy
            if self._cfg.learn.log_space:
                self._log_alpha = torch.log(torch.tensor([self._cfg.learn.alpha]))
                self._log_alpha = self._log_alpha.to('cuda' if self._cuda else 'cpu').requires_grad_(True)
                self._alpha_optim = torch.optim.Adam([self._log_alpha], lr=self._cfg.learn.learning_rate_alpha)
                self._alpha = self._log_alpha.detach().exp()
                self._auto_alpha = True
                self._log_space = True
            else:
                self._alpha = torch.tensor(
                    [self._cfg.learn.alpha], requires_grad=True, device='cuda' if self._cuda else 'cpu'
                )
                self._alpha_optim = torch.optim.Adam([self._alpha], lr=self._cfg.learn.learning_rate_alpha)
                self._auto_alpha = True
                self._log_space = False
        else:
            self._alpha = torch.tensor([self._cfg.learn.alpha])
            if self._cuda:
                self._alpha = self._alpha.cuda()
            self._auto_alpha = False
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_theta}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        self._target_model.train()
        self._forward_learn_cnt = 0

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=False,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        obs, next_obs = data['obs'], data['next_obs']
        reward = data['reward'].reshape(-1)
        done = data['done']
        weight = data['weight'] if data['weight'] is not None else 1.0

        # ---- critic update
        q_value = self._learn_model.forward({'obs': obs, 'action': data['action']}, mode='compute_critic')['q_value']
        with torch.no_grad():
            (mu, sigma) = self._learn_model.forward(next_obs, mode='compute_actor')['logit']
            dist = Independent(Normal(mu, sigma), 1)
            pred = dist.rsample()
            next_action = torch.tanh(pred)
            y = 1 - next_action.pow(2) + 1e-6
            next_log_prob = dist.log_prob(pred) - torch.log(y).sum(-1)
            next_q = self._target_model.forward({'obs': next_obs, 'action': next_action}, mode='compute_critic')['q_value']
            if self._twin_critic:
                next_q = torch.min(next_q[0], next_q[1])
            target_v = next_q - self._alpha * next_log_prob
            target_q = reward + self._gamma * (1 - done) * target_v
        if self._twin_critic:
            td1 = q_value[0] - target_q
            td2 = q_value[1] - target_q
            critic_loss = (td1.pow(2) * weight).mean() + (td2.pow(2) * weight).mean()
            td_error_per_sample = (td1.abs() + td2.abs()) / 2
        else:
            td = q_value - target_q
            critic_loss = (td.pow(2) * weight).mean()
            td_error_per_sample = td.abs()
        self._optimizer_q.zero_grad()
        critic_loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer_q.step()

        # ---- actor update
        (mu, sigma) = self._learn_model.forward(obs, mode='compute_actor')['logit']
        dist = Independent(Normal(mu, sigma), 1)
        pred = dist.rsample()
        action = torch.tanh(pred)
        y = 1 - action.pow(2) + 1e-6
        log_prob = dist.log_prob(pred) - torch.log(y).sum(-1)
        q = self._learn_model.forward({'obs': obs, 'action': action}, mode='compute_critic')['q_value']
        if self._twin_critic:
            q = torch.min(q[0], q[1])
        policy_loss = (self._alpha * log_prob - q).mean()
        self._optimizer_policy.zero_grad()
        policy_loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer_policy.step()

        # ---- temperature update
        alpha_loss = torch.zeros(())
        if self._auto_alpha:
            if self._log_space:
                alpha_loss = -(self._log_alpha * (log_prob.detach() + self._target_entropy)).mean()
                self._alpha_optim.zero_grad()
                alpha_loss.backward()
                self._alpha_optim.step()
                self._alpha = self._log_alpha.detach().exp()
            else:
                alpha_loss = -(self._alpha * (log_prob.detach() + self._target_entropy)).mean()
                self._alpha_optim.zero_grad()
                alpha_loss.backward()
                self._alpha_optim.step()
                with torch.no_grad():
                    self._alpha.clamp_(min=1e-8)
        self._forward_learn_cnt += 1
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr_q': self._optimizer_q.defaults['lr'],
            'cur_lr_p': self._optimizer_policy.defaults['lr'],
            'critic_loss': critic_loss.item(),
            'policy_loss': policy_loss.item(),
            'alpha_loss': alpha_loss.item() if isinstance(alpha_loss, torch.Tensor) else alpha_loss,
            'total_loss': critic_loss.item() + policy_loss.item(),
            'alpha': self._alpha.item() if isinstance(self._alpha, torch.Tensor) else float(self._alpha),
            'priority': td_error_per_sample.abs().tolist(),
        }

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr_q', 'cur_lr_p', 'critic_loss', 'policy_loss', 'alpha_loss', 'total_loss', 'alpha']

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._collect_model = model_wrap(self._model, wrapper_name='base')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            (mu, sigma) = self._collect_model.forward(collated, mode='compute_actor')['logit']
            dist = Independent(Normal(mu, sigma), 1)
            action = torch.tanh(dist.rsample())
            output = {'logit': (mu, sigma), 'action': action}
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs: Any, policy_output: Dict[str, Any], timestep: namedtuple) -> Dict[str, Any]:
        return {
            'obs': obs,
            'next_obs': timestep.obs,
            'action': policy_output['action'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        return get_train_sample(transitions, self._unroll_len)

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='base')
        self._eval_model.reset()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            (mu, sigma) = self._eval_model.forward(collated, mode='compute_actor')['logit']
            output = {'action': torch.tanh(mu)}
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}


@POLICY_REGISTRY.register('discrete_sac')
class DiscreteSACPolicy(SACPolicy):
    """Discrete-action SAC with categorical policy."""

    config = dict(
        type='discrete_sac',
        model=dict(twin_critic=True),
        learn=dict(
            update_per_collect=1,
            batch_size=256,
            learning_rate_q=3e-4,
            learning_rate_policy=3e-4,
            learning_rate_alpha=3e-4,
            target_theta=0.005,
            discount_factor=0.99,
            alpha=0.2,
            auto_alpha=True,
            log_space=True,
            target_entropy=None,
            ignore_done=False,
        ),
        other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000)),
    )

    def default_model(self) -> tuple:
        return 'discrete_qac', ['ding.model.template.qac']

    def _init_learn(self) -> None:
        super()._init_learn()
        if self._cfg.learn.auto_alpha and self._cfg.learn.target_entropy is None:
            action_shape = self._cfg.model.action_shape
            self._target_entropy = 0.98 * float(np.log(action_shape))

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(data, use_nstep=False, ignore_done=self._cfg.learn.ignore_done)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        obs, next_obs = data['obs'], data['next_obs']
        reward = data['reward'].reshape(-1)
        done = data['done']
        weight = data['weight'] if data['weight'] is not None else 1.0

        q_value = self._learn_model.forward(obs, mode='compute_critic')['q_value']
        with torch.no_grad():
            next_logit = self._learn_model.forward(next_obs, mode='compute_actor')['logit']
            next_pi = torch.softmax(next_logit, dim=-1)
            next_logpi = torch.log(next_pi + 1e-8)
            next_q = self._target_model.forward(next_obs, mode='compute_critic')['q_value']
            if self._twin_critic:
                next_q = torch.min(next_q[0], next_q[1])
            target_v = (next_pi * (next_q - self._alpha * next_logpi)).sum(-1)
            # multi-agent: target_v is [B, A] while reward/done are [B] —
            # broadcast per agent (shared team reward / episode done)
            while reward.dim() < target_v.dim():
                reward = reward.unsqueeze(-1)
            while done.dim() < target_v.dim():
                done = done.unsqueeze(-1)
            target_q = reward + self._gamma * (1 - done) * target_v
        act = data['action'].long()
        if self._twin_critic:
            q1 = q_value[0].gather(-1, act.unsqueeze(-1)).squeeze(-1)
            q2 = q_value[1].gather(-1, act.unsqueeze(-1)).squeeze(-1)
            td1, td2 = q1 - target_q, q2 - target_q
            critic_loss = (td1.pow(2) * weight).mean() + (td2.pow(2) * weight).mean()
            td_error_per_sample = (td1.abs() + td2.abs()) / 2
        else:
            q1 = q_value.gather(-1, act.unsqueeze(-1)).squeeze(-1)
            td1 = q1 - target_q
            critic_loss = (td1.pow(2) * weight).mean()
            td_error_per_sample = td1.abs()
        self._optimizer_q.zero_grad()
        critic_loss.backward()
        self._optimizer_q.step()

        logit = self._learn_model.forward(obs, mode='compute_actor')['logit']
        pi = torch.softmax(logit, dim=-1)
        logpi = torch.log(pi + 1e-8)
        with torch.no_grad():
            q = self._learn_model.forward(obs, mode='compute_critic')['q_value']
            if self._twin_critic:
                q = torch.min(q[0], q[1])
        policy_loss = (pi * (self._alpha * logpi - q)).sum(-1).mean()
        self._optimizer_policy.zero_grad()
        policy_loss.backward()
        self._optimizer_policy.step()

        alpha_loss = torch.zeros(())
        entropy = -(pi * logpi).sum(-1).mean()
        if self._auto_alpha:
            alpha_loss = (self._log_alpha * (entropy.detach() - self._target_entropy)).mean()
            self._alpha_optim.zero_grad()
            alpha_loss.backward()
            self._alpha_optim.step()
            self._alpha = self._log_alpha.detach().exp()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'critic_loss': critic_loss.item(),
            'policy_loss': policy_loss.item(),
            'alpha_loss': alpha_loss.item(),
            'total_loss': critic_loss.item() + policy_loss.item(),
            'alpha': self._alpha.item(),
            'entropy': entropy.item(),
            'priority': td_error_per_sample.abs().tolist(),
            'cur_lr': self._optimizer_q.defaults['lr'],
        }

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._collect_model = model_wrap(self._model, wrapper_name='eps_greedy_multinomial_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], eps: float = 0.0) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(collated, mode='compute_actor', eps=eps)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._eval_model.reset()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            output = self._eval_model.forward(collated, mode='compute_actor')
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}


@POLICY_REGISTRY.register('sqil_sac')
class SQILSACPolicy(SACPolicy):
    """SAC for SQIL imitation: trained on half expert (reward 1) / half agent
    (reward 0) batches assembled by the sqil entry."""

    config = dict(type='sqil_sac')
