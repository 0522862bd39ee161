"""ACER policy: off-policy actor-critic with retrace targets, truncated IS
and trust-region update.

Parity: reference ding/policy/acer.py ('acer').
"""
from collections import namedtuple
from typing import Any, Dict, List

import torch
import torch.nn.functional as F

from ding.model import model_wrap
from ding.rl_utils import (
    acer_policy_error, acer_value_error, acer_trust_region_update, compute_q_retraces, get_train_sample,
)
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate, timestep_collate
from .base_policy import Policy
import copy


@POLICY_REGISTRY.register('acer')
class ACERPolicy(Policy):

    config = dict(
        type='acer',
        cuda=False,
        on_policy=False,
        priority=False,
        unroll_len=32,
        model=dict(),
        learn=dict(
            update_per_collect=4,
            batch_size=16,
            learning_rate_actor=1e-4,
            learning_rate_critic=1e-4,
            c_clip_ratio=10.0,
            discount_factor=0.9,
            trust_region=True,
            trust_region_value=1.0,
            entropy_weight=0.0,
            target_theta=0.005,
        ),
        collect=dict(n_sample=16, collector=dict(type='sample')),
        eval=dict(),
        other=dict(replay_buffer=dict(replay_buffer_size=1000, ), ),
    )

    def default_model(self) -> tuple:
        return 'acer', ['ding.model.template.acer_model']

    def _init_learn(self) -> None:
        self._optimizer_actor = Adam(self._model.actor.parameters(), lr=self._cfg.learn.learning_rate_actor)
        self._optimizer_critic = Adam(self._model.critic.parameters(), lr=self._cfg.learn.learning_rate_critic)
        self._gamma = self._cfg.learn.discount_factor
        self._c_clip = self._cfg.learn.c_clip_ratio
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_theta}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        self._target_model.train()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = timestep_collate(data)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        T, B = data['action'].shape[:2]
        obs_flat = data['obs'].reshape(T * B, *data['obs'].shape[2:])
        actor_out = self._learn_model.forward(obs_flat, mode='compute_actor')['logit'].reshape(T, B, -1)
        critic_out = self._learn_model.forward(obs_flat, mode='compute_critic')['q_value'].reshape(T, B, -1)
        target_logit = F.log_softmax(actor_out, dim=-1)
        with torch.no_grad():
            behaviour_logit = F.log_softmax(data['logit'], dim=-1)
            ratio = torch.exp(target_logit - behaviour_logit)
            pi = torch.softmax(actor_out, dim=-1)
            v_pred = (pi * critic_out).sum(-1, keepdim=True)  # [T, B, 1]
            # bootstrap with an extra step copy
            q_values = torch.cat([critic_out, critic_out[-1:]], dim=0)
            v_all = torch.cat([v_pred, v_pred[-1:]], dim=0)
            reward = data['reward']
            if reward.dim() == 3:
                reward = reward.squeeze(-1)
            weights = 1 - data['done'].float()
            q_retraces = compute_q_retraces(q_values, v_all, reward, data['action'], weights, ratio, self._gamma)
        actor_loss, bc_loss = acer_policy_error(
            critic_out, q_retraces[:-1], v_pred, target_logit, data['action'], ratio, self._c_clip
        )
        dist = torch.distributions.Categorical(logits=actor_out)
        entropy_loss = dist.entropy().mean()
        total_actor_loss = (actor_loss + bc_loss).mean() - self._entropy_weight * entropy_loss
        if self._cfg.learn.trust_region:
            # trust-region projection on the logit gradient
            actor_grads = torch.autograd.grad(total_actor_loss, actor_out, retain_graph=True)
            with torch.no_grad():
                avg_logit = behaviour_logit
            updates = acer_trust_region_update(
                list(actor_grads), target_logit, avg_logit, self._cfg.learn.trust_region_value
            )
            self._optimizer_actor.zero_grad()
            actor_out.backward(updates[0], retain_graph=True)
            self._optimizer_actor.step()
        else:
            self._optimizer_actor.zero_grad()
            total_actor_loss.backward(retain_graph=True)
            self._optimizer_actor.step()
        critic_loss = acer_value_error(critic_out, q_retraces[:-1], data['action']).mean()
        self._optimizer_critic.zero_grad()
        critic_loss.backward()
        self._optimizer_critic.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer_actor.defaults['lr'],
            'actor_loss': total_actor_loss.item(),
            'critic_loss': critic_loss.item(),
            'total_loss': total_actor_loss.item() + critic_loss.item(),
            'entropy_loss': entropy_loss.item(),
        }

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'actor_loss', 'critic_loss', 'total_loss', 'entropy_loss']

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.unroll_len
        self._collect_model = model_wrap(self._model, wrapper_name='multinomial_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(collated, mode='compute_actor')
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs: Any, policy_output: Dict[str, Any], timestep: namedtuple) -> Dict[str, Any]:
        return {
            'obs': obs,
            'logit': policy_output['logit'],
            'action': policy_output['action'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        return get_train_sample(transitions, self._unroll_len)

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
