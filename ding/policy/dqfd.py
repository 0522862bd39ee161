"""DQfD, PDQN, D4PG, MADQN policies.

Parity: reference ding/policy/{dqfd,pdqn,d4pg,madqn}.py.
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import (
    dqfd_nstep_td_data, dqfd_nstep_td_error, get_nstep_return_data, get_train_sample, dist_nstep_td_data,
    dist_nstep_td_error, q_nstep_td_data, q_nstep_td_error,
)
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .common_utils import default_preprocess_learn
from .dqn import DQNPolicy
from .ddpg import DDPGPolicy


@POLICY_REGISTRY.register('dqfd')
class DQFDPolicy(DQNPolicy):
    """Deep Q-learning from demonstrations: n-step + 1-step TD + large-margin
    supervised loss on expert-labelled transitions."""

    config = dict(
        type='dqfd',
        priority=True,
        priority_IS_weight=True,
        nstep=10,
        learn=dict(
            update_per_collect=3,
            batch_size=64,
            learning_rate=0.001,
            target_update_freq=100,
            lambda1=1.0,   # n-step weight
            lambda2=1.0,   # supervised weight
            lambda_one_step_td=1.0,
            margin_function=0.8,
            per_train_iter_k=10,
            ignore_done=False,
        ),
    )

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        for d in data:
            d.setdefault('is_expert', 0)
            d.setdefault('done_one_step', d.get('done', False))
        collated = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        self._target_model.train()
        q_value = self._learn_model.forward(collated['obs'])['logit']
        with torch.no_grad():
            target_q = self._target_model.forward(collated['next_obs'])['logit']
            target_act = self._learn_model.forward(collated['next_obs'])['action']
            # one-step uses the first-step reward/next-obs approximation:
            # next_obs here is the n-step obs; reuse as one-step surrogate
            target_q_one = target_q
            target_act_one = target_act
        done_one = collated.get('done_one_step', collated['done'])
        if isinstance(done_one, torch.Tensor):
            done_one = done_one.float()
        td_data = dqfd_nstep_td_data(
            q_value, target_q, collated['action'], target_act, collated['reward'], collated['done'], done_one,
            collated['weight'], target_q_one, target_act_one, collated['is_expert']
        )
        loss, td_error_per_sample, parts = dqfd_nstep_td_error(
            td_data, self._gamma, self._cfg.learn.lambda1, self._cfg.learn.lambda2,
            self._cfg.learn.margin_function, self._cfg.learn.lambda_one_step_td, self._nstep,
            value_gamma=collated.get('value_gamma')
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
            'td1_loss': parts[0].item(),
            'tdn_loss': parts[1].item(),
            'supervised_loss': parts[2].item(),
            'priority': td_error_per_sample.abs().tolist(),
        }


@POLICY_REGISTRY.register('pdqn')
class PDQNPolicy(DQNPolicy):
    """Parameterized DQN for hybrid (discrete type + continuous args)
    actions: continuous net maximizes Q, discrete net does Q-learning over
    types given args."""

    config = dict(
        type='pdqn',
        priority=False,
        nstep=1,
        discount_factor=0.97,
        learn=dict(
            update_per_collect=3,
            batch_size=64,
            learning_rate_dis=0.001,
            learning_rate_cont=0.001,
            target_theta=0.005,
            update_circle=10,
            ignore_done=False,
        ),
        collect=dict(n_sample=8, unroll_len=1, noise_sigma=0.1),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
            replay_buffer=dict(replay_buffer_size=10000, ),
        ),
    )

    def default_model(self) -> tuple:
        return 'pdqn', ['ding.model.template.pdqn']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._optimizer_dis = Adam(self._model.dis_head.parameters(), lr=self._cfg.learn.learning_rate_dis)
        self._optimizer_cont = Adam(self._model.cont_head.parameters(), lr=self._cfg.learn.learning_rate_cont)
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_theta}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        self._target_model.train()
        self._forward_learn_cnt = 0

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(data, use_nstep=True, ignore_done=self._cfg.learn.ignore_done)
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        self._target_model.train()
        obs = collated['obs']
        action = collated['action']  # {'action_type': [B], 'action_args': [B, m]}
        # continuous branch: maximize Q(s, k, x_k)
        cont_out = self._learn_model.forward(obs, mode='compute_continuous')
        dis_out_c = self._learn_model.forward({'state': obs, 'action_args': cont_out['action_args']},
                                              mode='compute_discrete')
        cont_loss = -dis_out_c['logit'].sum(dim=-1).mean()
        self._optimizer_cont.zero_grad()
        cont_loss.backward()
        self._optimizer_cont.step()
        # discrete branch: q-learning with observed args
        dis_out = self._learn_model.forward({'state': obs, 'action_args': action['action_args']},
                                            mode='compute_discrete')
        q_value = dis_out['logit']
        with torch.no_grad():
            next_cont = self._target_model.forward(collated['next_obs'], mode='compute_continuous')
            next_dis = self._target_model.forward(
                {'state': collated['next_obs'], 'action_args': next_cont['action_args']}, mode='compute_discrete'
            )
            target_q = next_dis['logit']
            next_act = target_q.argmax(dim=-1)
        td_data = q_nstep_td_data(
            q_value, target_q, action['action_type'].long(), next_act, collated['reward'], collated['done'],
            collated['weight']
        )
        loss, td = q_nstep_td_error(td_data, self._gamma, nstep=self._nstep,
                                    value_gamma=collated.get('value_gamma'))
        self._optimizer_dis.zero_grad()
        loss.backward()
        self._optimizer_dis.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer_dis.defaults['lr'],
            'q_loss': loss.item(),
            'continuous_loss': cont_loss.item(),
            'total_loss': loss.item() + cont_loss.item(),
            'priority': td.abs().tolist(),
        }

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._collect_model = model_wrap(
            self._model, wrapper_name='hybrid_eps_greedy_multinomial_sample'
        )
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], eps: float) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(collated, eps=eps)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs, policy_output, timestep) -> Dict[str, Any]:
        return {
            'obs': obs,
            'next_obs': timestep.obs,
            'action': policy_output['action'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='hybrid_argmax_sample')
        self._eval_model.reset()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            output = self._eval_model.forward(collated)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}


@POLICY_REGISTRY.register('d4pg')
class D4PGPolicy(DDPGPolicy):
    """Distributed distributional DDPG: C51 critic + n-step returns."""

    config = dict(
        type='d4pg',
        nstep=3,
        priority=True,
        priority_IS_weight=True,
        model=dict(twin_critic=False, action_space='regression', v_min=-10, v_max=10, n_atom=51),
        learn=dict(
            update_per_collect=1,
            batch_size=256,
            learning_rate_actor=1e-3,
            learning_rate_critic=1e-3,
            ignore_done=False,
            target_theta=0.005,
            discount_factor=0.99,
            actor_update_freq=1,
            noise=False,
        ),
    )

    def default_model(self) -> tuple:
        return 'qac_dist', ['ding.model.template.qac_dist']

    def _init_learn(self) -> None:
        super()._init_learn()
        self._v_min = self._cfg.model.v_min
        self._v_max = self._cfg.model.v_max
        self._n_atom = self._cfg.model.n_atom
        self._nstep = self._cfg.nstep

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        collated = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight, use_nstep=True,
            ignore_done=self._cfg.learn.ignore_done
        )
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        self._target_model.train()
        out = self._learn_model.forward({'obs': collated['obs'], 'action': collated['action']},
                                        mode='compute_critic')
        q_dist = out['distribution']  # [B, n_atom]
        with torch.no_grad():
            next_action = self._target_model.forward(collated['next_obs'], mode='compute_actor')['action']
            target_out = self._target_model.forward(
                {'obs': collated['next_obs'], 'action': next_action}, mode='compute_critic'
            )
            target_dist = target_out['distribution']
        # categorical projection (per-sample single "action" slot)
        td_data = dist_nstep_td_data(
            q_dist.unsqueeze(1), target_dist.unsqueeze(1),
            torch.zeros(q_dist.shape[0], dtype=torch.long, device=q_dist.device),
            torch.zeros(q_dist.shape[0], dtype=torch.long, device=q_dist.device),
            collated['reward'], collated['done'], collated['weight']
        )
        critic_loss, td_error_per_sample = dist_nstep_td_error(
            td_data, self._gamma, self._v_min, self._v_max, self._n_atom, self._nstep,
            value_gamma=collated.get('value_gamma')
        )
        self._optimizer_critic.zero_grad()
        critic_loss.backward()
        self._optimizer_critic.step()
        # actor: maximize expected Q under the distributional critic
        actor_action = self._learn_model.forward(collated['obs'], mode='compute_actor')['action']
        actor_out = self._learn_model.forward({'obs': collated['obs'], 'action': actor_action},
                                              mode='compute_critic')
        actor_loss = -actor_out['q_value'].mean()
        self._optimizer_actor.zero_grad()
        actor_loss.backward()
        self._optimizer_actor.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr_actor': self._optimizer_actor.defaults['lr'],
            'cur_lr_critic': self._optimizer_critic.defaults['lr'],
            'critic_loss': critic_loss.item(),
            'actor_loss': actor_loss.item(),
            'total_loss': critic_loss.item() + actor_loss.item(),
            'priority': td_error_per_sample.abs().tolist(),
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        from collections import deque
        data = get_nstep_return_data(deque(transitions), self._nstep, gamma=self._gamma)
        return get_train_sample(list(data), self._unroll_len)


