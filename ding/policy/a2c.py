"""A2C policy. Parity: reference ding/policy/a2c.py."""
from collections import namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import a2c_data, a2c_error, a2c_error_continuous, get_gae_with_default_last_value, get_train_sample
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn


@POLICY_REGISTRY.register('a2c')
class A2CPolicy(Policy):

    config = dict(
        type='a2c',
        cuda=False,
        on_policy=True,
        priority=False,
        priority_IS_weight=False,
        action_space='discrete',
        transition_with_policy_data=True,
        model=dict(),
        learn=dict(
            batch_size=64,
            learning_rate=0.001,
            value_weight=0.5,
            entropy_weight=0.01,
            adv_norm=False,
            ignore_done=False,
            grad_norm=0.5,
        ),
        collect=dict(
            unroll_len=1,
            discount_factor=0.9,
            gae_lambda=0.95,
        ),
        eval=dict(),
    )

    def default_model(self) -> tuple:
        return 'vac', ['ding.model.template.vac']

    def _init_learn(self) -> None:
        self._optimizer = Adam(
            self._model.parameters(), lr=self._cfg.learn.learning_rate, grad_clip_type='clip_norm',
            clip_value=self._cfg.learn.grad_norm
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._action_space = self._cfg.action_space
        self._value_weight = self._cfg.learn.value_weight
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._adv_norm = self._cfg.learn.adv_norm
        self._gamma = self._cfg.collect.discount_factor
        self._gae_lambda = self._cfg.collect.gae_lambda
        self._learn_model.reset()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(data, ignore_done=self._cfg.learn.ignore_done, use_nstep=False)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        output = self._learn_model.forward(data['obs'], mode='compute_actor_critic')
        adv = data['adv']
        return_ = data['value'] + adv
        if self._adv_norm:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        error_fn = a2c_error_continuous if self._action_space == 'continuous' else a2c_error
        loss = error_fn(a2c_data(output['logit'], data['action'], output['value'], adv, return_, data['weight']))
        total_loss = loss.policy_loss + self._value_weight * loss.value_loss - self._entropy_weight * loss.entropy_loss
        self._optimizer.zero_grad()
        total_loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': total_loss.item(),
            'policy_loss': loss.policy_loss.item(),
            'value_loss': loss.value_loss.item(),
            'entropy_loss': loss.entropy_loss.item(),
        }

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'policy_loss', 'value_loss', 'entropy_loss']

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.collect.discount_factor
        self._gae_lambda = self._cfg.collect.gae_lambda
        if self._cfg.action_space == 'continuous':
            self._collect_model = model_wrap(self._model, wrapper_name='reparam_sample')
        else:
            self._collect_model = model_wrap(self._model, wrapper_name='multinomial_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(collated, mode='compute_actor_critic')
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs: Any, policy_output: Dict[str, Any], timestep: namedtuple) -> Dict[str, Any]:
        return {
            'obs': obs,
            'next_obs': timestep.obs,
            'action': policy_output['action'],
            'logit': policy_output['logit'],
            'value': policy_output['value'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        from collections import deque
        data = get_gae_with_default_last_value(
            deque(transitions), transitions[-1]['done'], self._gamma, self._gae_lambda, cuda=False
        )
        return get_train_sample(data, self._unroll_len)

    def _init_eval(self) -> None:
        if self._cfg.action_space == 'continuous':
            self._eval_model = model_wrap(self._model, wrapper_name='deterministic_sample')
        else:
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
