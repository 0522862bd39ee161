"""Offline-RL policies: BC (discrete/continuous), CQL, discrete CQL, TD3+BC,
IQL, EDAC.

Parity: reference ding/policy/{bc,cql,td3_bc,iql,edac}.py.
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import numpy as np
import torch
import torch.nn.functional as F
from torch.distributions import Independent, Normal

from ding.model import model_wrap
from ding.rl_utils import qrdqn_nstep_td_data, qrdqn_nstep_td_error
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn
from .sac import SACPolicy
from .ddpg import TD3Policy
from .c51 import QRDQNPolicy


@POLICY_REGISTRY.register('bc')
class BehaviourCloningPolicy(Policy):

    config = dict(
        type='bc',
        cuda=False,
        on_policy=False,
        continuous=False,
        action_shape=0,
        model=dict(),
        learn=dict(
            update_per_collect=1,
            batch_size=64,
            learning_rate=1e-3,
            lr_decay=False,
            momentum=0.9,
            weight_decay=1e-4,
            ce_label_smooth=False,
            show_accuracy=False,
            tanh_mask=False,
        ),
        collect=dict(unroll_len=1, ),
        eval=dict(),
        other=dict(replay_buffer=dict(replay_buffer_size=10000, )),
    )

    def default_model(self) -> tuple:
        if self._cfg.continuous:
            return 'continuous_bc', ['ding.model.template.bc']
        return 'bc', ['ding.model.template.bc']

    def _init_learn(self) -> None:
        self._optimizer = Adam(
            self._model.parameters(), lr=self._cfg.learn.learning_rate, weight_decay=self._cfg.learn.weight_decay
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        if self._cfg.continuous:
            self._loss = torch.nn.MSELoss()
        else:
            self._loss = torch.nn.CrossEntropyLoss()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(data)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        obs, action = data['obs'], data['action']
        out = self._learn_model.forward(obs)
        if self._cfg.continuous:
            pred = out['action'] if 'action' in out else out['logit']['mu']
            if action.dim() == pred.dim() - 1:
                action = action.unsqueeze(-1)
            loss = self._loss(pred, action)
        else:
            logit = out['logit']
            loss = self._loss(logit, action.long())
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        return {'cur_lr': self._optimizer.defaults['lr'], 'total_loss': loss.item()}

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        if self._cfg.continuous:
            self._collect_model = model_wrap(self._model, wrapper_name='base')
        else:
            self._collect_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(collated)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs, policy_output, timestep) -> Dict[str, Any]:
        return {
            'obs': obs, 'action': policy_output['action'], 'reward': timestep.reward, 'done': timestep.done,
            'next_obs': timestep.obs
        }

    def _get_train_sample(self, transitions):
        from ding.rl_utils import get_train_sample
        return get_train_sample(transitions, self._unroll_len)

    def _init_eval(self) -> None:
        self._init_collect()
        self._eval_model = self._collect_model

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        return self._forward_collect(data)


@POLICY_REGISTRY.register('cql')
class CQLPolicy(SACPolicy):
    """Conservative Q-learning on top of SAC."""

    config = dict(
        type='cql',
        learn=dict(
            update_per_collect=1,
            batch_size=256,
            learning_rate_q=3e-4,
            learning_rate_policy=1e-4,
            learning_rate_alpha=1e-4,
            target_theta=0.005,
            discount_factor=0.99,
            alpha=0.2,
            auto_alpha=True,
            log_space=True,
            min_q_weight=1.0,
            with_lagrange=False,
            lagrange_thresh=-1,
            num_actions=10,
            ignore_done=False,
            target_entropy=None,
        ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        self._min_q_weight = self._cfg.learn.min_q_weight
        self._num_actions = self._cfg.learn.num_actions

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(data, use_nstep=False, ignore_done=self._cfg.learn.ignore_done)
        if self._cuda:
            collated = to_device(collated, self._device)
        out = super()._forward_learn(data)
        # conservative penalty: push down Q on sampled actions, up on data actions
        obs, action = collated['obs'], collated['action']
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        B = obs.shape[0]
        rand_actions = torch.empty(B * self._num_actions, action.shape[-1], device=obs.device).uniform_(-1, 1)
        obs_rep = obs.repeat_interleave(self._num_actions, dim=0)
        q_rand = self._learn_model.forward({'obs': obs_rep, 'action': rand_actions}, mode='compute_critic')['q_value']
        if self._twin_critic:
            q_rand = [q.view(B, self._num_actions) for q in q_rand]
            q_data = self._learn_model.forward({'obs': obs, 'action': action}, mode='compute_critic')['q_value']
            cql_loss = sum(
                (torch.logsumexp(qr, dim=1) - qd).mean() for qr, qd in zip(q_rand, q_data)
            ) * self._min_q_weight
        else:
            q_rand = q_rand.view(B, self._num_actions)
            q_data = self._learn_model.forward({'obs': obs, 'action': action}, mode='compute_critic')['q_value']
            cql_loss = (torch.logsumexp(q_rand, dim=1) - q_data).mean() * self._min_q_weight
        self._optimizer_q.zero_grad()
        cql_loss.backward()
        self._optimizer_q.step()
        out['cql_loss'] = cql_loss.item()
        return out


@POLICY_REGISTRY.register('discrete_cql')
class DiscreteCQLPolicy(QRDQNPolicy):
    """Discrete CQL on top of QRDQN."""

    config = dict(
        type='discrete_cql',
        learn=dict(
            update_per_collect=1, batch_size=64, learning_rate=1e-4, target_update_freq=100, min_q_weight=1.0,
            ignore_done=False,
        ),
    )

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        self._target_model.train()
        output = self._learn_model.forward(collated['obs'])
        with torch.no_grad():
            target_output = self._target_model.forward(collated['next_obs'])
            target_act = self._learn_model.forward(collated['next_obs'])['logit'].argmax(dim=-1)
        td_data = qrdqn_nstep_td_data(
            output['q'], target_output['q'], collated['action'], target_act, collated['reward'], collated['done'],
            output['tau'], collated['weight']
        )
        loss, td_error_per_sample = qrdqn_nstep_td_error(
            td_data, self._gamma, self._nstep, value_gamma=collated.get('value_gamma')
        )
        # CQL term over mean-q (logit)
        q = output['logit']
        cql_loss = (torch.logsumexp(q, dim=1) - q.gather(1, collated['action'].unsqueeze(1)).squeeze(1)).mean()
        total = loss + self._cfg.learn.min_q_weight * cql_loss
        self._optimizer.zero_grad()
        total.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': total.item(),
            'cql_loss': cql_loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('td3_bc')
class TD3BCPolicy(TD3Policy):
    """TD3 + behaviour-cloning regularizer (offline)."""

    config = dict(
        type='td3_bc',
        learn=dict(
            update_per_collect=1, batch_size=256, learning_rate_actor=3e-4, learning_rate_critic=3e-4,
            ignore_done=False, target_theta=0.005, discount_factor=0.99, actor_update_freq=2, noise=True,
            noise_sigma=0.2, noise_range=dict(min=-0.5, max=0.5), alpha=2.5,
        ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        self._alpha_bc = self._cfg.learn.alpha

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(data, use_nstep=False, ignore_done=self._cfg.learn.ignore_done)
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        self._target_model.train()
        q_value = self._learn_model.forward(
            {'obs': collated['obs'], 'action': collated['action']}, mode='compute_critic'
        )['q_value']
        with torch.no_grad():
            next_action = self._target_model.forward(collated['next_obs'], mode='compute_actor')['action']
            target_q = self._target_model.forward(
                {'obs': collated['next_obs'], 'action': next_action}, mode='compute_critic'
            )['q_value']
            target_q = torch.min(target_q[0], target_q[1])
            reward = collated['reward'].reshape(-1)
            target = reward + self._gamma * (1 - collated['done']) * target_q
        td1 = q_value[0] - target
        td2 = q_value[1] - target
        critic_loss = td1.pow(2).mean() + td2.pow(2).mean()
        self._optimizer_critic.zero_grad()
        critic_loss.backward()
        self._optimizer_critic.step()
        actor_loss = torch.zeros(())
        bc_loss = torch.zeros(())
        if self._forward_learn_cnt % self._actor_update_freq == 0:
            pred_action = self._learn_model.forward(collated['obs'], mode='compute_actor')['action']
            q = self._learn_model.forward(
                {'obs': collated['obs'], 'action': pred_action}, mode='compute_critic'
            )['q_value'][0]
            lam = self._alpha_bc / q.abs().mean().detach()
            bc_loss = F.mse_loss(pred_action, collated['action'].reshape(pred_action.shape))
            actor_loss = -lam * q.mean() + bc_loss
            self._optimizer_actor.zero_grad()
            actor_loss.backward()
            self._optimizer_actor.step()
        self._forward_learn_cnt += 1
        self._target_model.update(self._learn_model.state_dict())
        return {
            'critic_loss': critic_loss.item(),
            'actor_loss': float(actor_loss.detach()),
            'bc_loss': float(bc_loss.detach()),
            'total_loss': critic_loss.item() + float(actor_loss.detach()),
            'cur_lr': self._optimizer_critic.defaults['lr'],
        }


@POLICY_REGISTRY.register('iql')
class IQLPolicy(SACPolicy):
    """Implicit Q-learning: expectile value regression + AWR policy
    extraction."""

    config = dict(
        type='iql',
        learn=dict(
            update_per_collect=1, batch_size=256, learning_rate_q=3e-4, learning_rate_policy=3e-4,
            learning_rate_alpha=3e-4, learning_rate_value=3e-4, target_theta=0.005, discount_factor=0.99,
            alpha=0.2, auto_alpha=False, log_space=True, expectile=0.7, beta=3.0, ignore_done=False,
            target_entropy=None,
        ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        # separate state-value net for expectile regression
        import torch.nn as nn
        obs_shape = self._cfg.model.obs_shape
        self._value_net = nn.Sequential(
            nn.Linear(obs_shape, 256), nn.ReLU(), nn.Linear(256, 256), nn.ReLU(), nn.Linear(256, 1)
        )
        if self._cuda:
            self._value_net.cuda()
        self._optimizer_value = Adam(self._value_net.parameters(), lr=self._cfg.learn.learning_rate_value)
        self._expectile = self._cfg.learn.expectile
        self._beta = self._cfg.learn.beta

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(data, use_nstep=False, ignore_done=self._cfg.learn.ignore_done)
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        self._target_model.train()
        obs, action = collated['obs'], collated['action']
        reward = collated['reward'].reshape(-1)
        done = collated['done']
        # value via expectile regression towards target Q
        with torch.no_grad():
            tq = self._target_model.forward({'obs': obs, 'action': action}, mode='compute_critic')['q_value']
            tq = torch.min(tq[0], tq[1]) if self._twin_critic else tq
        v = self._value_net(obs).squeeze(-1)
        diff = tq - v
        value_loss = (torch.where(diff > 0, self._expectile, 1 - self._expectile) * diff.pow(2)).mean()
        self._optimizer_value.zero_grad()
        value_loss.backward()
        self._optimizer_value.step()
        # critic: TD towards r + gamma V(s')
        with torch.no_grad():
            next_v = self._value_net(collated['next_obs']).squeeze(-1)
            target_q = reward + self._gamma * (1 - done) * next_v
        q_value = self._learn_model.forward({'obs': obs, 'action': action}, mode='compute_critic')['q_value']
        if self._twin_critic:
            critic_loss = (q_value[0] - target_q).pow(2).mean() + (q_value[1] - target_q).pow(2).mean()
        else:
            critic_loss = (q_value - target_q).pow(2).mean()
        self._optimizer_q.zero_grad()
        critic_loss.backward()
        self._optimizer_q.step()
        # AWR actor
        with torch.no_grad():
            adv = tq - v.detach()
            weight = torch.exp(self._beta * adv).clamp(max=100.0)
        (mu, sigma) = self._learn_model.forward(obs, mode='compute_actor')['logit']
        dist = Independent(Normal(mu, sigma), 1)
        # actions in data are env-space (tanh'ed); invert for log_prob stability
        raw = torch.atanh(action.clamp(-1 + 1e-6, 1 - 1e-6))
        log_prob = dist.log_prob(raw)
        policy_loss = -(weight * log_prob).mean()
        self._optimizer_policy.zero_grad()
        policy_loss.backward()
        self._optimizer_policy.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'value_loss': value_loss.item(),
            'critic_loss': critic_loss.item(),
            'policy_loss': policy_loss.item(),
            'total_loss': value_loss.item() + critic_loss.item() + policy_loss.item(),
            'cur_lr': self._optimizer_q.defaults['lr'],
        }


@POLICY_REGISTRY.register('edac')
class EDACPolicy(SACPolicy):
    """Ensemble-diversified actor critic (SAC-N + ensemble gradient
    diversity)."""

    config = dict(
        type='edac',
        model=dict(ensemble_num=10, ),
        learn=dict(
            update_per_collect=1, batch_size=256, learning_rate_q=3e-4, learning_rate_policy=3e-4,
            learning_rate_alpha=3e-4, target_theta=0.005, discount_factor=0.99, alpha=0.2, auto_alpha=True,
            log_space=True, eta=1.0, ignore_done=False, target_entropy=None,
        ),
    )

    def default_model(self) -> tuple:
        return 'edac', ['ding.model.template.bc']

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(data, use_nstep=False, ignore_done=self._cfg.learn.ignore_done)
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        self._target_model.train()
        obs, action = collated['obs'], collated['action']
        reward = collated['reward'].reshape(-1)
        done = collated['done']
        # q: [E, B]
        q_value = self._learn_model.forward({'obs': obs, 'action': action}, mode='compute_critic')['q_value']
        with torch.no_grad():
            (mu, sigma) = self._learn_model.forward(collated['next_obs'], mode='compute_actor')['logit']
            dist = Independent(Normal(mu, sigma), 1)
            pred = dist.rsample()
            next_action = torch.tanh(pred)
            y = 1 - next_action.pow(2) + 1e-6
            next_log_prob = dist.log_prob(pred) - torch.log(y).sum(-1)
            next_q = self._target_model.forward(
                {'obs': collated['next_obs'], 'action': next_action}, mode='compute_critic'
            )['q_value']
            target_q = next_q.min(dim=0)[0] - self._alpha * next_log_prob
            target = reward + self._gamma * (1 - done) * target_q
        critic_loss = (q_value - target.unsqueeze(0)).pow(2).mean()
        self._optimizer_q.zero_grad()
        critic_loss.backward()
        self._optimizer_q.step()
        # actor
        (mu, sigma) = self._learn_model.forward(obs, mode='compute_actor')['logit']
        dist = Independent(Normal(mu, sigma), 1)
        pred = dist.rsample()
        a = torch.tanh(pred)
        y = 1 - a.pow(2) + 1e-6
        log_prob = dist.log_prob(pred) - torch.log(y).sum(-1)
        q = self._learn_model.forward({'obs': obs, 'action': a}, mode='compute_critic')['q_value'].min(dim=0)[0]
        policy_loss = (self._alpha * log_prob - q).mean()
        self._optimizer_policy.zero_grad()
        policy_loss.backward()
        self._optimizer_policy.step()
        if self._auto_alpha:
            alpha_loss = -(self._log_alpha * (log_prob.detach() + self._target_entropy)).mean()
            self._alpha_optim.zero_grad()
            alpha_loss.backward()
            self._alpha_optim.step()
            self._alpha = self._log_alpha.detach().exp()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'critic_loss': critic_loss.item(),
            'policy_loss': policy_loss.item(),
            'total_loss': critic_loss.item() + policy_loss.item(),
            'cur_lr': self._optimizer_q.defaults['lr'],
        }
