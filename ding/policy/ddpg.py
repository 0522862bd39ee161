"""DDPG and TD3 policies.

Parity: reference ding/policy/ddpg.py ('ddpg' + 'td3' via twin_critic /
noise switches).
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import get_nstep_return_data, get_train_sample
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn


@POLICY_REGISTRY.register('ddpg')
class DDPGPolicy(Policy):

    config = dict(
        type='ddpg',
        cuda=False,
        on_policy=False,
        priority=False,
        priority_IS_weight=False,
        random_collect_size=25000,
        transition_with_policy_data=False,
        action_space='continuous',
        reward_batch_norm=False,
        multi_agent=False,
        model=dict(twin_critic=False, action_space='regression'),
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
            noise_sigma=0.2,
            noise_range=dict(min=-0.5, max=0.5),
        ),
        collect=dict(
            unroll_len=1,
            noise_sigma=0.1,
        ),
        eval=dict(),
        other=dict(replay_buffer=dict(replay_buffer_size=100000, ), ),
    )

    def default_model(self) -> tuple:
        return 'continuous_qac', ['ding.model.template.qac']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._twin_critic = self._cfg.model.twin_critic
        self._optimizer_actor = Adam(self._model.actor.parameters(), lr=self._cfg.learn.learning_rate_actor)
        self._optimizer_critic = Adam(self._model.critic.parameters(), lr=self._cfg.learn.learning_rate_critic)
        self._gamma = self._cfg.learn.discount_factor
        self._actor_update_freq = self._cfg.learn.actor_update_freq
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_theta}
        )
        if self._cfg.learn.noise:  # TD3 target policy smoothing
            self._target_model = model_wrap(
                self._target_model, wrapper_name='action_noise', noise_type='gauss',
                noise_kwargs={'mu': 0.0, 'sigma': self._cfg.learn.noise_sigma},
                noise_range=self._cfg.learn.noise_range
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
        # critic
        q_value = self._learn_model.forward({'obs': data['obs'], 'action': data['action']}, mode='compute_critic')['q_value']
        with torch.no_grad():
            next_actor_out = self._target_model.forward(data['next_obs'], mode='compute_actor')
            if 'action' in next_actor_out:
                next_action = next_actor_out['action']
            else:  # hybrid actor: compose greedy type + args
                next_action = {
                    'action_type': next_actor_out['logit'].argmax(-1),
                    'action_args': next_actor_out['action_args'],
                }
            next_data = {'obs': data['next_obs'], 'action': next_action}
            target_q = self._target_model.forward(next_data, mode='compute_critic')['q_value']
        reward = data['reward']
        if reward.dim() > 1:
            reward = reward.reshape(-1)
        if self._twin_critic:
            target_q = torch.min(target_q[0], target_q[1])
            target = reward + self._gamma * (1 - data['done']) * target_q
            td1 = q_value[0] - target.detach()
            td2 = q_value[1] - target.detach()
            weight = data['weight'] if data['weight'] is not None else torch.ones_like(td1)
            critic_loss = (td1.pow(2) * weight).mean() + (td2.pow(2) * weight).mean()
            td_error_per_sample = (td1.abs() + td2.abs()) / 2
        else:
            target = reward + self._gamma * (1 - data['done']) * target_q
            td = q_value - target.detach()
            weight = data['weight'] if data['weight'] is not None else torch.ones_like(td)
            critic_loss = (td.pow(2) * weight).mean()
            td_error_per_sample = td.abs()
        self._optimizer_critic.zero_grad()
        critic_loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer_critic.step()
        # delayed actor update
        actor_loss = torch.zeros(())
        if self._forward_learn_cnt % self._actor_update_freq == 0:
            actor_out = self._learn_model.forward(data['obs'], mode='compute_actor')
            if 'action' in actor_out:
                actor_data = {'obs': data['obs'], 'action': actor_out['action']}
            else:  # hybrid: soft type probs would break the critic cat; use logit passthrough
                actor_data = {
                    'obs': data['obs'], 'logit': actor_out['logit'],
                    'action': {'action_type': actor_out['logit'].argmax(-1),
                               'action_args': actor_out['action_args']},
                }
            q = self._learn_model.forward(actor_data, mode='compute_critic')['q_value']
            if self._twin_critic:
                q = q[0]
            actor_loss = -q.mean()
            self._optimizer_actor.zero_grad()
            actor_loss.backward()
            if self._cfg.multi_gpu:
                self.sync_gradients(self._model)
            self._optimizer_actor.step()
        self._forward_learn_cnt += 1
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr_actor': self._optimizer_actor.defaults['lr'],
            'cur_lr_critic': self._optimizer_critic.defaults['lr'],
            'critic_loss': critic_loss.item(),
            'actor_loss': actor_loss.item(),
            'total_loss': critic_loss.item() + actor_loss.item(),
            'q_value': (q_value[0] if self._twin_critic else q_value).mean().item(),
            'priority': td_error_per_sample.abs().tolist(),
        }

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr_actor', 'cur_lr_critic', 'critic_loss', 'actor_loss', 'total_loss', 'q_value']

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        if self._cfg.get('action_space', None) == 'hybrid':  # PADDPG lane
            self._collect_model = model_wrap(self._model, wrapper_name='hybrid_eps_greedy_multinomial_sample')
        else:
            self._collect_model = model_wrap(
                self._model, wrapper_name='action_noise', noise_type='gauss',
                noise_kwargs={'mu': 0.0, 'sigma': self._cfg.collect.noise_sigma}, noise_range=None
            )
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
            'next_obs': timestep.obs,
            'action': policy_output['action'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        return get_train_sample(transitions, self._unroll_len)

    def _init_eval(self) -> None:
        if self._cfg.get('action_space', None) == 'hybrid':
            self._eval_model = model_wrap(self._model, wrapper_name='hybrid_argmax_sample')
        else:
            self._eval_model = model_wrap(self._model, wrapper_name='base')
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


@POLICY_REGISTRY.register('td3')
class TD3Policy(DDPGPolicy):
    """TD3 = DDPG + twin critic + delayed actor + target policy smoothing."""

    config = dict(
        type='td3',
        model=dict(twin_critic=True, action_space='regression'),
        learn=dict(
            update_per_collect=1,
            batch_size=256,
            learning_rate_actor=1e-3,
            learning_rate_critic=1e-3,
            ignore_done=False,
            target_theta=0.005,
            discount_factor=0.99,
            actor_update_freq=2,
            noise=True,
            noise_sigma=0.2,
            noise_range=dict(min=-0.5, max=0.5),
        ),
    )
