"""C51 / QRDQN / IQN / FQF / Rainbow / SQL / MDQN / BDQ / SQN policies —
the distributional & soft value-based family, sharing DQNPolicy's
collect/eval machinery.

Parity: reference ding/policy/{c51,qrdqn,iqn,fqf,rainbow,sql,mdqn,bdq,sqn}.py.
"""
import copy
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import (
    dist_nstep_td_data, dist_nstep_td_error, qrdqn_nstep_td_data, qrdqn_nstep_td_error, iqn_nstep_td_data,
    iqn_nstep_td_error, fqf_nstep_td_data, fqf_nstep_td_error, fqf_calculate_fraction_loss, q_nstep_td_data,
    q_nstep_sql_td_error, m_q_1step_td_data, m_q_1step_td_error, bdq_nstep_td_error, q_nstep_td_error,
)
from ding.torch_utils import Adam, to_device, RMSprop
from ding.utils import POLICY_REGISTRY
from .common_utils import default_preprocess_learn
from .dqn import DQNPolicy


@POLICY_REGISTRY.register('c51')
class C51Policy(DQNPolicy):

    config = dict(
        type='c51',
        model=dict(v_min=-10, v_max=10, n_atom=51),
        learn=dict(
            update_per_collect=3,
            batch_size=64,
            learning_rate=0.001,
            target_update_freq=100,
            ignore_done=False,
        ),
    )

    def default_model(self) -> tuple:
        return 'c51dqn', ['ding.model.template.q_learning']

    def _init_learn(self) -> None:
        super()._init_learn()
        self._v_min = self._cfg.model.v_min
        self._v_max = self._cfg.model.v_max
        self._n_atom = self._cfg.model.n_atom

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        output = self._learn_model.forward(data['obs'])
        with torch.no_grad():
            target_output = self._target_model.forward(data['next_obs'])
            target_act = self._learn_model.forward(data['next_obs'])['logit'].argmax(dim=-1)
        td_data = dist_nstep_td_data(
            output['distribution'], target_output['distribution'], data['action'], target_act, data['reward'],
            data['done'], data['weight']
        )
        loss, td_error_per_sample = dist_nstep_td_error(
            td_data, self._gamma, self._v_min, self._v_max, self._n_atom, self._nstep,
            value_gamma=data.get('value_gamma')
        )
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
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('qrdqn')
class QRDQNPolicy(DQNPolicy):

    config = dict(
        type='qrdqn',
        learn=dict(update_per_collect=3, batch_size=64, learning_rate=0.001, target_update_freq=100,
                   ignore_done=False),
    )

    def default_model(self) -> tuple:
        return 'qrdqn', ['ding.model.template.q_learning']

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        output = self._learn_model.forward(data['obs'])
        with torch.no_grad():
            target_output = self._target_model.forward(data['next_obs'])
            target_act = self._learn_model.forward(data['next_obs'])['logit'].argmax(dim=-1)
        td_data = qrdqn_nstep_td_data(
            output['q'], target_output['q'], data['action'], target_act, data['reward'], data['done'],
            output['tau'], data['weight']
        )
        loss, td_error_per_sample = qrdqn_nstep_td_error(
            td_data, self._gamma, self._nstep, value_gamma=data.get('value_gamma')
        )
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('iqn')
class IQNPolicy(DQNPolicy):

    config = dict(
        type='iqn',
        learn=dict(update_per_collect=3, batch_size=64, learning_rate=0.001, target_update_freq=100,
                   kappa=1.0, ignore_done=False),
    )

    def default_model(self) -> tuple:
        return 'iqn', ['ding.model.template.q_learning']

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        output = self._learn_model.forward(data['obs'])
        with torch.no_grad():
            target_output = self._target_model.forward(data['next_obs'])
            target_act = self._learn_model.forward(data['next_obs'])['logit'].argmax(dim=-1)
        td_data = iqn_nstep_td_data(
            output['q'], target_output['q'], data['action'], target_act, data['reward'], data['done'],
            output['quantiles'].view(-1, data['action'].shape[0], 1), data['weight']
        )
        loss, td_error_per_sample = iqn_nstep_td_error(
            td_data, self._gamma, self._nstep, kappa=self._cfg.learn.kappa, value_gamma=data.get('value_gamma')
        )
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('fqf')
class FQFPolicy(DQNPolicy):

    config = dict(
        type='fqf',
        learn=dict(
            update_per_collect=3, batch_size=64, learning_rate_fraction=2.5e-9, learning_rate_quantile=0.00005,
            target_update_freq=100, kappa=1.0, ent_coef=0, ignore_done=False,
        ),
    )

    def default_model(self) -> tuple:
        return 'fqf', ['ding.model.template.q_learning']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        fraction_params = list(self._model.head.fqf_fc.parameters())
        fraction_ids = {id(p) for p in fraction_params}
        quantile_params = [p for p in self._model.parameters() if id(p) not in fraction_ids]
        self._fraction_loss_optimizer = RMSprop(
            fraction_params, lr=self._cfg.learn.learning_rate_fraction, alpha=0.95, eps=0.00001
        )
        self._quantile_loss_optimizer = Adam(quantile_params, lr=self._cfg.learn.learning_rate_quantile)
        self._optimizer = self._quantile_loss_optimizer
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='assign',
            update_kwargs={'freq': self._cfg.learn.target_update_freq}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._learn_model.train()
        self._target_model.train()
        self._forward_learn_cnt = 0

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        output = self._learn_model.forward(data['obs'])
        with torch.no_grad():
            target_output = self._target_model.forward(data['next_obs'])
            target_act = self._learn_model.forward(data['next_obs'])['logit'].argmax(dim=-1)
        td_data = fqf_nstep_td_data(
            output['q'], target_output['q'], data['action'], target_act, data['reward'], data['done'],
            output['quantiles_hats'], data['weight']
        )
        loss, td_error_per_sample = fqf_nstep_td_error(
            td_data, self._gamma, self._nstep, kappa=self._cfg.learn.kappa, value_gamma=data.get('value_gamma')
        )
        fraction_loss = fqf_calculate_fraction_loss(
            output['q_tau_i'], output['q'], output['quantiles'], data['action']
        )
        ent = output['entropies'].mean()
        fraction_loss = fraction_loss - self._cfg.learn.ent_coef * ent
        self._fraction_loss_optimizer.zero_grad()
        fraction_loss.backward(retain_graph=True)
        self._quantile_loss_optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._fraction_loss_optimizer.step()
        self._quantile_loss_optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._quantile_loss_optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'fraction_loss': fraction_loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('rainbow')
class RainbowDQNPolicy(C51Policy):
    """Rainbow: noisy nets (reset per forward) + dueling-distributional head
    + n-step + PER."""

    config = dict(
        type='rainbow',
        priority=True,
        priority_IS_weight=True,
        model=dict(v_min=-10, v_max=10, n_atom=51),
        nstep=3,
    )

    def default_model(self) -> tuple:
        return 'rainbowdqn', ['ding.model.template.q_learning']

    def _reset_noise(self, model):
        from ding.torch_utils import NoisyLinearLayer
        for m in model.modules():
            if isinstance(m, NoisyLinearLayer):
                m.reset_noise()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        self._reset_noise(self._model)
        self._reset_noise(self._target_model.model)
        return super()._forward_learn(data)

    def _forward_collect(self, data: Dict[int, Any], eps: float = -1) -> Dict[int, Any]:
        self._reset_noise(self._model)
        return super()._forward_collect(data, eps=-1 if eps is None else eps)


@POLICY_REGISTRY.register('sql')
class SQLPolicy(DQNPolicy):
    """Soft Q-learning (energy-based)."""

    config = dict(
        type='sql',
        learn=dict(update_per_collect=3, batch_size=64, learning_rate=0.001, target_update_freq=100, alpha=0.12,
                   ignore_done=False),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        self._alpha = self._cfg.learn.alpha

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        q_value = self._learn_model.forward(data['obs'])['logit']
        with torch.no_grad():
            target_q_value = self._target_model.forward(data['next_obs'])['logit']
            target_act = target_q_value.argmax(dim=-1)
        td_data = q_nstep_td_data(
            q_value, target_q_value, data['action'], target_act, data['reward'], data['done'], data['weight']
        )
        loss, td_error_per_sample = q_nstep_sql_td_error(
            td_data, self._gamma, self._alpha, self._nstep, value_gamma=data.get('value_gamma')
        )
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._collect_model = model_wrap(self._model, wrapper_name='eps_greedy_multinomial_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], eps: float) -> Dict[int, Any]:
        from ding.utils.data import default_collate, default_decollate
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(collated, eps=eps, alpha=self._cfg.learn.alpha)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}


@POLICY_REGISTRY.register('mdqn')
class MDQNPolicy(DQNPolicy):
    """Munchausen DQN."""

    config = dict(
        type='mdqn',
        entropy_tau=0.03,
        m_alpha=0.9,
        learn=dict(update_per_collect=3, batch_size=64, learning_rate=0.001, target_update_freq=100,
                   ignore_done=False),
    )

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=False,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        q_value = self._learn_model.forward(data['obs'])['logit']
        with torch.no_grad():
            target_q_current = self._target_model.forward(data['obs'])['logit']
            target_q_next = self._target_model.forward(data['next_obs'])['logit']
        td_data = m_q_1step_td_data(
            q_value, target_q_current, target_q_next, data['action'], data['reward'].reshape(-1), data['done'],
            data['weight']
        )
        loss, td_error_per_sample, action_gap = m_q_1step_td_error(
            td_data, self._gamma, self._cfg.entropy_tau, self._cfg.m_alpha
        )
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('bdq')
class BDQPolicy(DQNPolicy):
    """Branching dueling Q for discretized high-dim action spaces."""

    config = dict(
        type='bdq',
        learn=dict(update_per_collect=3, batch_size=64, learning_rate=0.001, target_update_freq=100,
                   ignore_done=False),
    )

    def default_model(self) -> tuple:
        return 'bdq', ['ding.model.template.q_learning']

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        q_value = self._learn_model.forward(data['obs'])['logit']  # [B, D, bins]
        with torch.no_grad():
            target_q_value = self._target_model.forward(data['next_obs'])['logit']
            target_act = self._learn_model.forward(data['next_obs'])['logit'].argmax(dim=-1)
        action = data['action']
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        td_data = q_nstep_td_data(
            q_value, target_q_value, action, target_act, data['reward'], data['done'], data['weight']
        )
        loss, td_error_per_sample = bdq_nstep_td_error(
            td_data, self._gamma, self._nstep, value_gamma=data.get('value_gamma')
        )
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('sqn')
class SQNPolicy(DQNPolicy):
    """Soft Q network with auto temperature (discrete SAC-style Q-learning)."""

    config = dict(
        type='sqn',
        learn=dict(
            update_per_collect=3, batch_size=64, learning_rate_q=0.001, learning_rate_alpha=0.0003,
            target_update_freq=0, target_theta=0.005, alpha=0.2, auto_alpha=True, ignore_done=False,
        ),
    )

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate_q)
        action_shape = self._cfg.model.action_shape
        self._target_entropy = 0.98 * float(torch.log(torch.tensor(float(action_shape))))
        self._log_alpha = torch.log(torch.tensor(self._cfg.learn.alpha, dtype=torch.float32))
        self._log_alpha = self._log_alpha.to('cuda' if self._cuda else 'cpu').requires_grad_(True)
        self._alpha_optim = torch.optim.Adam([self._log_alpha], lr=self._cfg.learn.learning_rate_alpha)
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_theta}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._learn_model.train()
        self._target_model.train()
        self._forward_learn_cnt = 0

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(data, use_nstep=False, ignore_done=self._cfg.learn.ignore_done)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        alpha = self._log_alpha.exp()
        q_value = self._learn_model.forward(data['obs'])['logit']
        q_a = q_value.gather(-1, data['action'].unsqueeze(-1)).squeeze(-1)
        with torch.no_grad():
            next_q = self._target_model.forward(data['next_obs'])['logit']
            pi = torch.softmax(next_q / alpha, dim=-1)
            target_v = (pi * (next_q - alpha * torch.log(pi + 1e-8))).sum(-1)
            reward = data['reward']
            if reward.dim() > 1:
                reward = reward.reshape(-1)
            target = reward + self._gamma * (1 - data['done']) * target_v
        td_error = (q_a - target)
        loss = (td_error.pow(2) * (data['weight'] if data['weight'] is not None else 1)).mean()
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        # temperature update
        with torch.no_grad():
            cur_pi = torch.softmax(q_value / alpha.detach(), dim=-1)
            entropy = -(cur_pi * torch.log(cur_pi + 1e-8)).sum(-1).mean()
        alpha_loss = self._log_alpha * (entropy.detach() - self._target_entropy)
        self._alpha_optim.zero_grad()
        alpha_loss.backward()
        self._alpha_optim.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'alpha': alpha.item(),
            'entropy': entropy.item(),
            'priority': td_error.abs().tolist(),
        }

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._collect_model = model_wrap(self._model, wrapper_name='eps_greedy_multinomial_sample')
        self._collect_model.reset()
