"""IMPALA policy: v-trace off-policy actor-critic over unrolled trajectories.

Parity: reference ding/policy/impala.py ('impala'). The v-trace reverse scan
runs in the HIP kernel lane on GPU (ding/ops scan kernels).
"""
from collections import namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import vtrace_data, vtrace_error_discrete_action, get_train_sample
from ding.torch_utils import Adam, RMSprop, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate, timestep_collate
from .base_policy import Policy


@POLICY_REGISTRY.register('impala')
class IMPALAPolicy(Policy):

    config = dict(
        type='impala',
        cuda=False,
        on_policy=False,
        priority=False,
        priority_IS_weight=False,
        unroll_len=32,
        transition_with_policy_data=True,
        action_space='discrete',
        model=dict(),
        learn=dict(
            update_per_collect=4,
            batch_size=16,
            learning_rate=0.0005,
            grad_clip_type=None,
            clip_value=None,
            optim='adam',
            value_weight=0.5,
            entropy_weight=0.0001,
            discount_factor=0.99,
            lambda_=0.95,
            rho_clip_ratio=1.0,
            c_clip_ratio=1.0,
            rho_pg_clip_ratio=1.0,
            ignore_done=False,
        ),
        collect=dict(n_sample=16, collector=dict(type='sample', ), ),
        eval=dict(),
        other=dict(replay_buffer=dict(replay_buffer_size=1000, ), ),
    )

    def default_model(self) -> tuple:
        return 'vac', ['ding.model.template.vac']

    def _init_learn(self) -> None:
        self._action_space = self._cfg.action_space
        optim_type = self._cfg.learn.optim
        if optim_type == 'rmsprop':
            self._optimizer = RMSprop(self._model.parameters(), lr=self._cfg.learn.learning_rate)
        else:
            self._optimizer = Adam(
                self._model.parameters(), lr=self._cfg.learn.learning_rate,
                grad_clip_type=self._cfg.learn.grad_clip_type, clip_value=self._cfg.learn.clip_value,
                flatten_grads=self._cfg.learn.get('flatten_grads', self._cuda and not self._cfg.multi_gpu),
            )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._value_weight = self._cfg.learn.value_weight
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._gamma = self._cfg.learn.discount_factor
        self._lambda = self._cfg.learn.lambda_
        self._rho_clip_ratio = self._cfg.learn.rho_clip_ratio
        self._c_clip_ratio = self._cfg.learn.c_clip_ratio
        self._rho_pg_clip_ratio = self._cfg.learn.rho_pg_clip_ratio
        # hipGraph capture of the whole learn step (fwd + v-trace + bwd):
        # static [T, B] shapes make IMPALA a one-graph-per-step workload
        self._cuda_graph = self._cfg.learn.get('cuda_graph', False) and not self._cfg.multi_gpu
        self._bf16 = self._cfg.learn.get('bf16', False)
        self._graphed_step = None
        self._learn_model.reset()

    def _data_preprocess_learn(self, data):
        """Collate unrolled samples -> time-major tensors [T, B, ...].

        Accepts either a list of per-sample dicts (the collector path) or an
        already-collated time-major dict whose tensors may live on the GPU
        (the same-node trajectory fast path / bench path) — the latter skips
        host collation entirely.
        """
        if not isinstance(data, dict):
            data = timestep_collate(data)
        if self._cuda:
            data = to_device(data, self._device)
        data['weight'] = data.get('weight', None)
        if 'obs_plus_1' not in data:
            data['obs_plus_1'] = torch.cat([data['obs'], data['next_obs'][-1:]], dim=0).float()
        return data

    def _learn_step(self, data: Dict[str, Any]) -> Dict[str, Any]:
        """fwd + v-trace + bwd on already-device-resident [T(+1), B] tensors;
        sync-free (returns 0-dim GPU tensors) so it can be hipGraph-captured."""
        T, B = data['done'].shape[:2]
        obs_flat = data['obs_plus_1'].reshape(-1, *data['obs_plus_1'].shape[2:])
        if self._bf16:
            # opt-in bf16 lane: conv/GEMM fwd+bwd in bf16, fp32 master
            # weights, v-trace loss math in fp32
            with torch.autocast(obs_flat.device.type, dtype=torch.bfloat16):
                output = self._learn_model.forward(obs_flat, mode='compute_actor_critic')
            output = {k: v.float() for k, v in output.items()}
        else:
            output = self._learn_model.forward(obs_flat, mode='compute_actor_critic')
        target_logit = output['logit'].reshape(T + 1, B, -1)[:-1]
        value = output['value'].reshape(T + 1, B)
        rewards = data['reward']
        if rewards.dim() == 3:
            rewards = rewards.squeeze(-1)
        vt_data = vtrace_data(target_logit, data['logit'], data['action'], value, rewards, data.get('weight'))
        loss = vtrace_error_discrete_action(
            vt_data, self._gamma, self._lambda, self._rho_clip_ratio, self._c_clip_ratio, self._rho_pg_clip_ratio
        )
        total_loss = loss.policy_loss + self._value_weight * loss.value_loss \
            - self._entropy_weight * loss.entropy_loss
        self._optimizer.zero_grad(set_to_none=False)
        total_loss.backward()
        return {
            'total_loss': total_loss.detach(),
            'policy_loss': loss.policy_loss.detach(),
            'value_loss': loss.value_loss.detach(),
            'entropy_loss': loss.entropy_loss.detach(),
        }

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = self._data_preprocess_learn(data)
        self._learn_model.train()
        use_graph = (
            self._cuda_graph and isinstance(data.get('obs_plus_1'), torch.Tensor)
            and data['obs_plus_1'].is_cuda and data.get('weight') is None
        )
        if use_graph:
            if self._graphed_step is None:
                from ding.torch_utils.hip_graph import GraphedStep
                self._graphed_step = GraphedStep(self._learn_step)
            inputs = {k: v for k, v in data.items() if isinstance(v, torch.Tensor)}
            out = self._graphed_step(inputs)
            self._optimizer.step()
            losses = {k: float(v) for k, v in out.items()}
            return {'cur_lr': self._optimizer.defaults['lr'], **losses}
        out = self._learn_step(data)
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            **{k: v.item() for k, v in out.items()},
        }

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'policy_loss', 'value_loss', 'entropy_loss']

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
            'next_obs': timestep.obs,
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


@POLICY_REGISTRY.register('pg')
class PGPolicy(Policy):
    """Vanilla policy gradient (REINFORCE) with MC returns."""

    config = dict(
        type='pg',
        cuda=False,
        on_policy=True,
        action_space='discrete',
        model=dict(),
        learn=dict(
            batch_size=64,
            learning_rate=0.001,
            entropy_weight=0.01,
            grad_norm=5,
            ignore_done=False,
        ),
        collect=dict(
            unroll_len=1,
            discount_factor=0.99,
            collector=dict(get_train_sample=True, type='episode'),
        ),
        eval=dict(),
    )

    def default_model(self) -> tuple:
        return 'pg', ['ding.model.template.pg']

    def _init_learn(self) -> None:
        self._optimizer = Adam(
            self._model.parameters(), lr=self._cfg.learn.learning_rate, grad_clip_type='clip_norm',
            clip_value=self._cfg.learn.grad_norm
        )
        self._learn_model = self._model
        self._entropy_weight = self._cfg.learn.entropy_weight

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        from .common_utils import default_preprocess_learn
        data = default_preprocess_learn(data, use_nstep=False)
        if self._cuda:
            data = to_device(data, self._device)
        self._model.train()
        output = self._model(data['obs'])
        return_ = data['return']
        dist = torch.distributions.Categorical(logits=output['logit']) if self._cfg.action_space == 'discrete' \
            else torch.distributions.Independent(
                torch.distributions.Normal(output['logit']['mu'], output['logit']['sigma']), 1)
        log_prob = dist.log_prob(data['action'])
        policy_loss = -(log_prob * return_).mean()
        entropy_loss = dist.entropy().mean()
        total_loss = policy_loss - self._entropy_weight * entropy_loss
        self._optimizer.zero_grad()
        total_loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': total_loss.item(),
            'policy_loss': policy_loss.item(),
            'entropy_loss': entropy_loss.item(),
        }

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.collect.discount_factor
        if self._cfg.action_space == 'discrete':
            self._collect_model = model_wrap(self._model, wrapper_name='multinomial_sample')
        else:
            self._collect_model = model_wrap(self._model, wrapper_name='reparam_sample')
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

    def _process_transition(self, obs: Any, policy_output: Dict[str, Any], timestep: namedtuple) -> Dict[str, Any]:
        return {
            'obs': obs,
            'action': policy_output['action'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        # discounted MC return per episode
        R = 0.0
        for i in reversed(range(len(transitions))):
            R = self._gamma * R + float(transitions[i]['reward'].item())
            transitions[i]['return'] = torch.tensor([R])
        from ding.rl_utils import get_train_sample
        return get_train_sample(transitions, self._unroll_len)

    def _init_eval(self) -> None:
        if self._cfg.action_space == 'discrete':
            self._eval_model = model_wrap(self._model, wrapper_name='argmax_sample')
        else:
            self._eval_model = model_wrap(self._model, wrapper_name='deterministic_sample')
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
