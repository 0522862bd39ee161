"""DQN policy (double DQN + dueling + n-step + PER support).

Parity: reference ding/policy/dqn.py ('dqn' registration; _forward_learn:224,
checkpoint format {'model','target_model','optimizer'} per :327-334).
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List, Optional

import torch

from ding.model import model_wrap
from ding.rl_utils import q_nstep_td_data, q_nstep_td_error, get_nstep_return_data, get_train_sample
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn


@POLICY_REGISTRY.register('dqn')
class DQNPolicy(Policy):

    config = dict(
        type='dqn',
        cuda=False,
        on_policy=False,
        priority=False,
        priority_IS_weight=False,
        discount_factor=0.97,
        nstep=1,
        model=dict(),
        learn=dict(
            update_per_collect=3,
            batch_size=64,
            learning_rate=0.001,
            target_update_freq=100,
            target_theta=0.005,
            ignore_done=False,
        ),
        collect=dict(
            n_sample=8,
            unroll_len=1,
        ),
        eval=dict(),
        other=dict(
            eps=dict(
                type='exp',
                start=0.95,
                end=0.1,
                decay=10000,
            ),
            replay_buffer=dict(replay_buffer_size=10000, ),
        ),
    )

    def default_model(self) -> tuple:
        return 'dqn', ['ding.model.template.q_learning']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate)
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        # target network: periodic assign by default
        if self._cfg.learn.get('target_update_freq', None):
            self._target_model = model_wrap(
                copy.deepcopy(self._model), wrapper_name='target', update_type='assign',
                update_kwargs={'freq': self._cfg.learn.target_update_freq}
            )
        else:
            self._target_model = model_wrap(
                copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
                update_kwargs={'theta': self._cfg.learn.target_theta}
            )
        self._learn_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._learn_model.train()
        self._target_model.train()
        self._forward_learn_cnt = 0

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data,
            use_priority=self._priority,
            use_priority_IS_weight=self._cfg.priority_IS_weight,
            use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done,
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        # opt-in bf16 lane: net fwd in autocast-bf16 (fp32 master weights,
        # fp32 TD/loss math, no loss scaling)
        import contextlib
        amp = torch.autocast(data['obs'].device.type, dtype=torch.bfloat16) \
            if self._cfg.learn.get('bf16', False) else contextlib.nullcontext()
        with torch.no_grad(), amp:
            target_q_value = self._target_model.forward(data['next_obs'])['logit'].float()
            target_q_action = self._learn_model.forward(data['next_obs'])['action']  # double DQN
        with amp:
            q_value = self._learn_model.forward(data['obs'])['logit']
        q_value = q_value.float()
        value_gamma = data.get('value_gamma')
        td_data = q_nstep_td_data(
            q_value, target_q_value, data['action'], target_q_action, data['reward'], data['done'], data['weight']
        )
        loss, td_error_per_sample = q_nstep_td_error(td_data, self._gamma, nstep=self._nstep, value_gamma=value_gamma)
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        self._forward_learn_cnt += 1
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'q_value': q_value.mean().item(),
            'priority': td_error_per_sample.abs().tolist(),
            'target_q_value': target_q_value.mean().item(),
        }

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'q_value', 'target_q_value']

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._collect_model = model_wrap(self._model, wrapper_name='eps_greedy_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], eps: float) -> Dict[int, Any]:
        data_id = list(data.keys())
        data = default_collate(list(data.values()))
        if self._cuda:
            data = to_device(data, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(data, eps=eps)
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
        transitions = get_nstep_return_data(transitions, self._nstep, gamma=self._gamma)
        return get_train_sample(list(transitions), self._unroll_len)

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._eval_model.reset()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        data = default_collate(list(data.values()))
        if self._cuda:
            data = to_device(data, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            output = self._eval_model.forward(data)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}


@POLICY_REGISTRY.register('dqn_stdim')
class DQNSTDIMPolicy(DQNPolicy):
    """DQN + ST-DIM auxiliary contrastive representation loss.

    Parity: reference ding/policy/dqn.py DQNSTDIMPolicy.
    """

    config = dict(
        type='dqn_stdim',
        aux_loss_weight=0.003,
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        from ding.torch_utils.loss import ContrastiveLoss
        obs_shape = self._cfg.model.obs_shape
        x_size, y_size = self._get_encoding_size()
        self._aux_model = ContrastiveLoss(x_size, y_size)
        if self._cuda:
            self._aux_model.cuda()
        self._aux_optimizer = Adam(self._aux_model.parameters(), lr=self._cfg.learn.learning_rate)
        self._aux_loss_weight = self._cfg.aux_loss_weight

    def _get_encoding_size(self):
        obs = self._cfg.model.obs_shape
        if isinstance(obs, int):
            test = torch.randn(1, obs)
        else:
            test = torch.randn(1, *obs)
        if self._cuda:
            test = test.cuda()
        with torch.no_grad():
            x = self._model.encoder(test)
        return x.shape[1], x.shape[1]

    def _aux_encode(self, data):
        with torch.no_grad():
            x = self._model.encoder(data['obs'])
            y = self._model.encoder(data['next_obs'])
        return x, y

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            collated = to_device(collated, self._device)
        x, y = self._aux_encode(collated)
        aux_loss = self._aux_model(x, y) * self._aux_loss_weight
        self._aux_optimizer.zero_grad()
        aux_loss.backward()
        self._aux_optimizer.step()
        out = super()._forward_learn(data)
        out['aux_loss'] = aux_loss.item()
        return out

    def _monitor_vars_learn(self) -> List[str]:
        return super()._monitor_vars_learn() + ['aux_loss']
