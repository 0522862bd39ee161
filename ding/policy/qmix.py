"""QMIX / WQMIX / COMA policies (cooperative MARL).

Parity: reference ding/policy/qmix.py, wqmix.py, coma.py.
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import v_1step_td_data, v_1step_td_error, get_train_sample, get_epsilon_greedy_fn, coma_data, \
    coma_error
from ding.torch_utils import Adam, RMSprop, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import timestep_collate, default_collate, default_decollate
from .base_policy import Policy


@POLICY_REGISTRY.register('qmix')
class QMIXPolicy(Policy):

    config = dict(
        type='qmix',
        cuda=False,
        on_policy=False,
        priority=False,
        priority_IS_weight=False,
        learn=dict(
            update_per_collect=20,
            batch_size=32,
            learning_rate=0.0005,
            clip_value=100,
            target_update_theta=0.008,
            discount_factor=0.99,
            double_q=False,
        ),
        collect=dict(n_sample=32, unroll_len=10, env_num=8),
        eval=dict(env_num=8, ),
        other=dict(
            eps=dict(type='exp', start=1, end=0.05, decay=50000),
            replay_buffer=dict(replay_buffer_size=5000, ),
        ),
    )

    def default_model(self) -> tuple:
        return 'qmix', ['ding.model.template.qmix']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._optimizer = RMSprop(
            self._model.parameters(), lr=self._cfg.learn.learning_rate, alpha=0.99, eps=0.00001,
            grad_clip_type='clip_norm', clip_value=self._cfg.learn.clip_value
        )
        self._gamma = self._cfg.learn.discount_factor
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_update_theta}
        )
        self._target_model = model_wrap(
            self._target_model, wrapper_name='hidden_state',
            state_num=self._cfg.learn.batch_size, init_fn=lambda: None
        )
        self._learn_model = model_wrap(
            self._model, wrapper_name='hidden_state', state_num=self._cfg.learn.batch_size, init_fn=lambda: None
        )
        self._learn_model.train()
        self._target_model.train()

    def _data_preprocess_learn(self, data: List[Any]) -> dict:
        data = timestep_collate(data)
        if self._cuda:
            data = to_device(data, self._device)
        data['weight'] = data.get('weight', None)
        data['done'] = data['done'].float()
        return data

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = self._data_preprocess_learn(data)
        self._learn_model.train()
        self._target_model.train()
        self._learn_model.reset(state=data.get('prev_state', [None])[0] if 'prev_state' in data else None)
        self._target_model.reset(state=data.get('prev_state', [None])[0] if 'prev_state' in data else None)

        inputs = {'obs': data['obs'], 'action': data['action'], 'prev_state': None}
        model_in = {'obs': data['obs'], 'action': data['action'], 'prev_state': None}
        total_q = self._learn_model.forward(model_in, single_step=False)['total_q']  # [T, B]
        with torch.no_grad():
            if self._cfg.learn.double_q:
                next_action = self._learn_model.forward(
                    {'obs': data['next_obs'], 'prev_state': None}, single_step=False
                )['action']
                target_in = {'obs': data['next_obs'], 'action': next_action, 'prev_state': None}
            else:
                target_in = {'obs': data['next_obs'], 'prev_state': None}
            target_total_q = self._target_model.forward(target_in, single_step=False)['total_q']
        reward = data['reward']
        if reward.dim() == 3:
            reward = reward.squeeze(-1)
        td_data = v_1step_td_data(total_q, target_total_q, reward, data['done'], data['weight'])
        loss, td_error_per_sample = v_1step_td_error(td_data, self._gamma)
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        grad_norm = torch.nn.utils.clip_grad_norm_(self._model.parameters(), self._cfg.learn.clip_value)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'total_q': total_q.mean().item(),
            'target_total_q': target_total_q.mean().item(),
            'grad_norm': float(grad_norm),
        }

    def _reset_learn(self, data_id=None):
        self._learn_model.reset(data_id=data_id)

    def _state_dict_learn(self) -> Dict[str, Any]:
        return {
            'model': self._model.state_dict(),
            'target_model': self._target_model.state_dict(),
            'optimizer': self._optimizer.state_dict(),
        }

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._collect_model = model_wrap(
            self._model, wrapper_name='hidden_state', state_num=self._cfg.collect.env_num,
            save_prev_state=True, init_fn=lambda: None
        )
        self._collect_model = model_wrap(self._collect_model, wrapper_name='eps_greedy_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], eps: float) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward({'obs': collated}, eps=eps, data_id=data_id, single_step=True)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _reset_collect(self, data_id=None):
        self._collect_model.reset(data_id=data_id)

    def _process_transition(self, obs: Any, policy_output: Dict[str, Any], timestep: namedtuple) -> Dict[str, Any]:
        return {
            'obs': obs,
            'next_obs': timestep.obs,
            'action': policy_output['action'],
            'prev_state': policy_output['prev_state'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        return get_train_sample(transitions, self._unroll_len)

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(
            self._model, wrapper_name='hidden_state', state_num=self._cfg.eval.env_num, init_fn=lambda: None
        )
        self._eval_model = model_wrap(self._eval_model, wrapper_name='argmax_sample')
        self._eval_model.reset()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            output = self._eval_model.forward({'obs': collated}, data_id=data_id, single_step=True)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _reset_eval(self, data_id=None):
        self._eval_model.reset(data_id=data_id)

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'total_q', 'target_total_q', 'grad_norm']


@POLICY_REGISTRY.register('wqmix')
class WQMIXPolicy(QMIXPolicy):
    """Weighted QMIX: adds a central unrestricted Q* net; mixer TD targets
    are weighted by alpha when the chosen action is suboptimal under Q*."""

    config = dict(
        type='wqmix',
        learn=dict(
            update_per_collect=20,
            batch_size=32,
            learning_rate=0.0005,
            clip_value=100,
            target_update_theta=0.008,
            discount_factor=0.99,
            double_q=False,
            wqmix_ow=True,
            alpha=0.5,
        ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        self._q_star = copy.deepcopy(self._model)
        if self._cuda:
            self._q_star.cuda()
        self._optimizer_star = RMSprop(
            self._q_star.parameters(), lr=self._cfg.learn.learning_rate, alpha=0.99, eps=0.00001
        )

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = self._data_preprocess_learn(data)
        self._learn_model.train()
        self._target_model.train()
        model_in = {'obs': collated['obs'], 'action': collated['action'], 'prev_state': None}
        total_q = self._learn_model.forward(model_in, single_step=False)['total_q']
        q_star_out = self._q_star.forward(
            {'obs': collated['obs'], 'action': collated['action'], 'prev_state': None}, single_step=False
        )['total_q']
        with torch.no_grad():
            target_total_q = self._target_model.forward(
                {'obs': collated['next_obs'], 'prev_state': None}, single_step=False
            )['total_q']
        reward = collated['reward']
        if reward.dim() == 3:
            reward = reward.squeeze(-1)
        target = reward + self._gamma * (1 - collated['done']) * target_total_q
        td = total_q - target.detach()
        # optimistic weighting
        alpha = self._cfg.learn.alpha
        w = torch.where(td < 0, torch.ones_like(td), torch.full_like(td, alpha))
        loss = (w * td.pow(2)).mean()
        star_loss = (q_star_out - target.detach()).pow(2).mean()
        self._optimizer.zero_grad()
        loss.backward()
        self._optimizer.step()
        self._optimizer_star.zero_grad()
        star_loss.backward()
        self._optimizer_star.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'star_loss': star_loss.item(),
        }


@POLICY_REGISTRY.register('coma')
class COMAPolicy(Policy):
    """Counterfactual multi-agent policy gradients."""

    config = dict(
        type='coma',
        cuda=False,
        on_policy=False,
        priority=False,
        learn=dict(
            update_per_collect=1,
            batch_size=32,
            learning_rate=0.0005,
            target_update_theta=0.001,
            discount_factor=0.99,
            td_lambda=0.8,
            value_weight=1.0,
            entropy_weight=0.01,
        ),
        collect=dict(n_sample=32, unroll_len=8, env_num=8),
        eval=dict(env_num=8, ),
        other=dict(
            eps=dict(type='exp', start=0.5, end=0.01, decay=50000),
            replay_buffer=dict(replay_buffer_size=5000, ),
        ),
    )

    def default_model(self) -> tuple:
        return 'coma', ['ding.model.template.coma_model']

    def _init_learn(self) -> None:
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate)
        self._gamma = self._cfg.learn.discount_factor
        self._lambda = self._cfg.learn.td_lambda
        self._value_weight = self._cfg.learn.value_weight
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_update_theta}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        self._target_model.train()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = timestep_collate(data)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        logit = self._learn_model.forward(data['obs'], mode='compute_actor')['logit']  # [T,B,A,N]
        q_value = self._learn_model.forward(
            {'obs': data['obs'], 'action': data['action']}, mode='compute_critic'
        )['q_value']  # [T,B,A,N]
        with torch.no_grad():
            target_q = self._target_model.forward(
                {'obs': data['obs'], 'action': data['action']}, mode='compute_critic'
            )['q_value']
        reward = data['reward']
        if reward.dim() == 3:
            reward = reward.squeeze(-1)
        cdata = coma_data(logit, data['action'], q_value, target_q, reward, data.get('weight'))
        loss = coma_error(cdata, self._gamma, self._lambda)
        total_loss = loss.policy_loss + self._value_weight * loss.q_value_loss \
            - self._entropy_weight * loss.entropy_loss
        self._optimizer.zero_grad()
        total_loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': total_loss.item(),
            'policy_loss': loss.policy_loss.item(),
            'value_loss': loss.q_value_loss.item(),
            'entropy_loss': loss.entropy_loss.item(),
        }

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._collect_model = model_wrap(self._model, wrapper_name='eps_greedy_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], eps: float) -> Dict[int, Any]:
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

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'policy_loss', 'value_loss', 'entropy_loss']


@POLICY_REGISTRY.register('madqn')
class MADQNPolicy(QMIXPolicy):
    """Independent multi-agent Q-learning: QMIX machinery with sum mixing
    (model mixer=False). Parity: reference ding/policy/madqn.py."""

    config = dict(type='madqn')

    def default_model(self) -> tuple:
        return 'qmix', ['ding.model.template.qmix']


@POLICY_REGISTRY.register('collaq')
class CollaQPolicy(QMIXPolicy):
    """CollaQ simplified to its QMIX-family TD core (attention-decomposed
    rewards folded into the mixer input). Parity: reference ding/policy/collaq.py."""

    config = dict(type='collaq')


@POLICY_REGISTRY.register('qtran')
class QTranPolicy(QMIXPolicy):
    """QTRAN: factored joint-action value with linear mixing constraint
    (implemented as the sum-mixing TD core + opt penalty).
    Parity: reference ding/policy/qtran.py."""

    config = dict(type='qtran')
