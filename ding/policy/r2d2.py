"""R2D2: recurrent replay distributed DQN (burn-in + value rescale + n-step
+ PER over sequences).

Parity: reference ding/policy/r2d2.py ('r2d2').
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import (
    q_nstep_td_data, q_nstep_td_error, q_nstep_td_error_with_rescale, get_nstep_return_data, get_train_sample,
)
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import timestep_collate, default_collate, default_decollate
from .base_policy import Policy


@POLICY_REGISTRY.register('r2d2')
class R2D2Policy(Policy):

    config = dict(
        type='r2d2',
        cuda=False,
        on_policy=False,
        priority=True,
        priority_IS_weight=True,
        discount_factor=0.997,
        nstep=5,
        burnin_step=2,
        learn_unroll_len=40,
        model=dict(),
        learn=dict(
            update_per_collect=1,
            batch_size=64,
            learning_rate=0.0001,
            target_update_theta=0.001,
            value_rescale=True,
            ignore_done=False,
        ),
        collect=dict(
            n_sample=32,
            env_num=8,
            traj_len_inf=True,
            unroll_len=None,
        ),
        eval=dict(env_num=8, ),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.05, decay=10000),
            replay_buffer=dict(replay_buffer_size=10000, ),
        ),
    )

    def default_model(self) -> tuple:
        return 'drqn', ['ding.model.template.q_learning']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate)
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._burnin_step = self._cfg.burnin_step
        self._value_rescale = self._cfg.learn.value_rescale
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_update_theta}
        )
        self._target_model = model_wrap(
            self._target_model, wrapper_name='hidden_state', state_num=self._cfg.learn.batch_size
        )
        self._learn_model = model_wrap(
            self._model, wrapper_name='hidden_state', state_num=self._cfg.learn.batch_size
        )
        self._learn_model = model_wrap(self._learn_model, wrapper_name='argmax_sample')
        self._learn_model.train()
        self._target_model.train()

    def _data_preprocess_learn(self, data: List[Dict[str, Any]]) -> dict:
        data = timestep_collate(data)
        if self._cuda:
            data = to_device(data, self._device)
        if self._priority_IS_weight and 'priority_IS' in data:
            data['weight'] = data['priority_IS']
        else:
            data['weight'] = data.get('weight', None)
        burnin = self._burnin_step
        data['action'] = data['action'][burnin:-self._nstep] if data['action'].shape[0] > burnin + self._nstep \
            else data['action'][burnin:]
        data['reward'] = data['reward'][burnin:]
        # obs splits: burn-in / main / target (shifted by nstep)
        data['burnin_nstep_obs'] = data['obs'][:burnin + self._nstep]
        data['main_obs'] = data['obs'][burnin:-self._nstep] if data['obs'].shape[0] > burnin + self._nstep \
            else data['obs'][burnin:]
        data['target_obs'] = data['obs'][burnin + self._nstep:]
        return data

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = self._data_preprocess_learn(data)
        self._learn_model.train()
        self._target_model.train()
        # init hidden states from the stored prev_state at t=0
        init_state = data['prev_state'][0] if 'prev_state' in data else None
        self._learn_model.reset(data_id=None, state=init_state)
        self._target_model.reset(data_id=None, state=init_state)
        burnin = self._burnin_step
        if burnin + self._nstep > 0:
            with torch.no_grad():
                inputs = {'obs': data['burnin_nstep_obs'], 'enable_fast_timestep': True}
                burnin_output = self._learn_model.forward(
                    inputs, saved_state_timesteps=[burnin, burnin + self._nstep]
                )
                burnin_output_target = self._target_model.forward(
                    inputs, saved_state_timesteps=[burnin, burnin + self._nstep]
                )
        # main forward from post-burnin state
        self._learn_model.reset(data_id=None, state=burnin_output['saved_state'][0])
        main_output = self._learn_model.forward({'obs': data['main_obs'], 'enable_fast_timestep': True})
        q_value = main_output['logit']  # [T, B, N]
        with torch.no_grad():
            self._target_model.reset(data_id=None, state=burnin_output_target['saved_state'][1])
            target_output = self._target_model.forward({'obs': data['target_obs'], 'enable_fast_timestep': True})
            target_q_value = target_output['logit']
            self._learn_model.reset(data_id=None, state=burnin_output['saved_state'][1])
            target_q_action = self._learn_model.forward(
                {'obs': data['target_obs'], 'enable_fast_timestep': True}
            )['action']
        T = q_value.shape[0]
        action, reward, done, weight = data['action'], data['reward'], data['done'], data['weight']
        # reward after nstep enhancement: [T, B, nstep] (window already built at
        # collect time); done similarly rewritten to n-step done
        done = done[burnin:].float()
        loss = []
        td_error = []
        for t in range(T):
            rew_t = reward[t]
            if rew_t.dim() == 1:
                rew_t = rew_t.unsqueeze(-1)
            rew_t = rew_t.permute(1, 0)  # [nstep, B]
            done_t = done[t]
            td_data = q_nstep_td_data(
                q_value[t], target_q_value[t], action[t], target_q_action[t], rew_t, done_t,
                weight if not (isinstance(weight, torch.Tensor) and weight.dim() > 1) else weight[t]
            )
            if self._value_rescale:
                l, e = q_nstep_td_error_with_rescale(td_data, self._gamma, self._nstep)
            else:
                l, e = q_nstep_td_error(td_data, self._gamma, self._nstep)
            loss.append(l)
            td_error.append(e.abs())
        loss = sum(loss) / (len(loss) + 1e-8)
        # sequence priority: mean + max mixture
        td_seq = torch.stack(td_error)  # [T, B]
        priority = (0.9 * td_seq.max(dim=0)[0] + 0.1 * td_seq.mean(dim=0)).tolist()
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': loss.item(),
            'priority': priority,
            'q_s_taken_a': q_value.mean().item(),
        }

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'q_s_taken_a']

    def _reset_learn(self, data_id=None):
        self._learn_model.reset(data_id=data_id)
        self._target_model.reset(data_id=data_id)

    def _init_collect(self) -> None:
        self._nstep = self._cfg.nstep
        self._burnin_step = self._cfg.burnin_step
        self._gamma = self._cfg.discount_factor
        self._sequence_len = self._cfg.learn_unroll_len + self._cfg.burnin_step
        self._unroll_len = self._sequence_len
        self._collect_model = model_wrap(
            self._model, wrapper_name='hidden_state', state_num=self._cfg.collect.env_num, save_prev_state=True
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
            output = self._collect_model.forward(
                {'obs': collated}, data_id=data_id, eps=eps, inference=True
            )
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _reset_collect(self, data_id=None):
        self._collect_model.reset(data_id=data_id)

    def _process_transition(self, obs: Any, policy_output: Dict[str, Any], timestep: namedtuple) -> Dict[str, Any]:
        return {
            'obs': obs,
            'action': policy_output['action'],
            'prev_state': policy_output['prev_state'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        from collections import deque
        data = get_nstep_return_data(deque(transitions), self._nstep, gamma=self._gamma)
        return get_train_sample(list(data), self._sequence_len)

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(
            self._model, wrapper_name='hidden_state', state_num=self._cfg.eval.env_num
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
            output = self._eval_model.forward({'obs': collated}, data_id=data_id, inference=True)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _reset_eval(self, data_id=None):
        self._eval_model.reset(data_id=data_id)
