"""PPO policies: on-policy PPO (discrete/continuous/hybrid, multi-agent
aware), PPO-PG (no critic), off-policy PPO.

Parity: reference ding/policy/ppo.py ('ppo', 'ppo_pg', 'ppo_offpolicy',
'ppo_stdim' registrations, ~1,900 LoC).
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from ding.model import model_wrap
from ding.rl_utils import (
    ppo_data, ppo_error, ppo_error_continuous, ppo_policy_data, ppo_policy_error, gae, gae_data, get_gae,
    get_train_sample, ppo_policy_error_continuous, get_nstep_return_data,
)
from ding.torch_utils import Adam, to_device
from ding.utils import split_data_generator
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn


@POLICY_REGISTRY.register('ppo')
class PPOPolicy(Policy):
    """On-policy PPO with GAE, value/dual clip, recompute-adv epochs."""

    config = dict(
        type='ppo',
        cuda=False,
        on_policy=True,
        priority=False,
        priority_IS_weight=False,
        recompute_adv=True,
        action_space='discrete',
        nstep_return=False,
        multi_agent=False,
        transition_with_policy_data=True,
        model=dict(),
        learn=dict(
            epoch_per_collect=10,
            batch_size=64,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
            ppo_param_init=True,
            grad_clip_type='clip_norm',
            grad_clip_value=0.5,
            ignore_done=False,
        ),
        collect=dict(
            unroll_len=1,
            discount_factor=0.99,
            gae_lambda=0.95,
        ),
        eval=dict(),
    )

    def default_model(self) -> tuple:
        if self._cfg.multi_agent:
            return 'mavac', ['ding.model.template.mavac']
        return 'vac', ['ding.model.template.vac']

    def _init_learn(self) -> None:
        self._action_space = self._cfg.action_space
        assert self._action_space in ('discrete', 'continuous', 'hybrid')
        if self._cfg.learn.ppo_param_init:
            for n, m in self._model.named_modules():
                if isinstance(m, torch.nn.Linear):
                    torch.nn.init.orthogonal_(m.weight, gain=np.sqrt(2))
                    torch.nn.init.zeros_(m.bias)
            if self._action_space in ('continuous', 'hybrid'):
                for m in getattr(self._model.actor_head, 'modules', lambda: [])():
                    if isinstance(m, torch.nn.Linear):
                        torch.nn.init.zeros_(m.bias)
                        m.weight.data.copy_(0.01 * m.weight.data)

        # NOTE: capturable Adam inside the graph was A/B'd on MI355X: no
        # throughput gain (219 vs 214 ms/step) and it broke graph-vs-eager
        # equivalence; the optimizer step stays eager after replay.
        _capturable = False
        self._optimizer = Adam(
            self._model.parameters(),
            lr=self._cfg.learn.learning_rate,
            grad_clip_type=self._cfg.learn.grad_clip_type,
            clip_value=self._cfg.learn.grad_clip_value,
            capturable=_capturable,
            # flat grad views: 1-kernel clip/zero_grad (rocprof: per-param
            # reduce/fill storm ~25 ms/step); DDP keeps per-param grads for
            # the bucketed reducer's own packing
            flatten_grads=self._cfg.learn.get('flatten_grads', self._cuda and not self._cfg.multi_gpu),
        )
        self._optimizer_capturable = _capturable
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._value_weight = self._cfg.learn.value_weight
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._clip_ratio = self._cfg.learn.clip_ratio
        self._adv_norm = self._cfg.learn.adv_norm
        self._value_norm = self._cfg.learn.value_norm
        if self._value_norm:
            from ding.utils import RunningMeanStd
            self._running_mean_std = RunningMeanStd(epsilon=1e-4, shape=(1, ))
        self._gamma = self._cfg.collect.discount_factor
        self._gae_lambda = self._cfg.collect.gae_lambda
        self._recompute_adv = self._cfg.recompute_adv
        # opt-in bf16 learner lane: model fwd (convs/GEMMs) runs under
        # autocast-bf16 with fp32 master weights and fp32 loss math — bf16
        # needs no loss scaling. Works inside hipGraph capture too (the
        # autocast weight-cast kernels are captured and replayed).
        self._bf16 = self._cfg.learn.get('bf16', False)
        # hipGraph capture of the minibatch step (MI355X: the step is
        # launch-bound — see ding/torch_utils/hip_graph.py). Single-process
        # only: the bucketed DDP reducer's backward hooks are not replayed
        # by graphs, so multi_gpu keeps the eager path.
        self._cuda_graph = self._cfg.learn.get('cuda_graph', False) and not self._cfg.multi_gpu
        # NHWC conv layout (MI355X: MIOpen's channels-last solvers measured
        # ~16% faster on the Atari stack); inputs are converted on entry
        self._channels_last = self._cfg.learn.get('channels_last', False) and self._cuda
        if self._channels_last:
            self._model.to(memory_format=torch.channels_last)
        self._graphed_step = None
        self._values_graph = None
        self._graph_info_keys = None
        self._learn_model.reset()

    def _amp_ctx(self, ref: torch.Tensor):
        """autocast-bf16 context for the opt-in bf16 lane (nullcontext when
        the lane is off); fp32 master weights, no loss scaling."""
        import contextlib
        if not self._bf16 or not isinstance(ref, torch.Tensor):
            return contextlib.nullcontext()
        return torch.autocast(ref.device.type, dtype=torch.bfloat16)

    def _graphed_values(self, both: torch.Tensor, chunk: int, fresh: bool = True) -> torch.Tensor:
        """hipGraph-captured no-grad chunked critic pass for recompute-adv.
        The [obs; next_obs] tensor is identical across the epoch loop, so
        the 722 MB static-input copy happens once per _forward_learn
        (``fresh`` is True only on epoch 0); later epochs replay copy-free.
        The chunk size is baked in at first capture."""
        if getattr(self, '_values_graph', None) is None:
            from ding.torch_utils.hip_graph import GraphedStep

            def fn(inp):
                with torch.no_grad(), self._amp_ctx(inp['both']):
                    chunks = torch.split(inp['both'], chunk, dim=0)
                    return {
                        'values': torch.cat(
                            [self._learn_model.forward(c, mode='compute_critic')['value'] for c in chunks], dim=0
                        ).float()
                    }

            self._values_graph = GraphedStep(fn)
        g = self._values_graph
        key = g._shape_key({'both': both})
        if not fresh and g._graph is not None and key == g._key:
            # same tensor as the epoch-0 copy (caller tracks freshness —
            # id() comparison would be unsafe across allocator reuse)
            g._graph.replay()
            return g._static_out['values'].clone()
        return g({'both': both})['values'].clone()

    def _graphed_minibatch(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        """Replay (or first capture) the fwd+loss+bwd hipGraph for one
        minibatch; optimizer step stays eager (grad-clip Adam is not
        capture-safe). Returns 0-dim GPU tensors — read them lazily."""
        if self._graphed_step is None:
            from ding.torch_utils.hip_graph import GraphedStep
            from ding.ops import dispatch as _dispatch
            wv, we = self._value_weight, self._entropy_weight

            def step_fn(b):
                with self._amp_ctx(b['obs']):
                    output = self._learn_model.forward(b['obs'], mode='compute_actor_critic')
                if self._bf16:
                    output = {k: v.float() for k, v in output.items()}
                adv = b['adv']
                if self._adv_norm:
                    adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                policy_loss, value_loss, entropy_loss, approx_kl, clipfrac = _dispatch.fused_ppo_error(
                    output['logit'], b['logit'], b['action'], output['value'], b['value'],
                    adv, b['return'], None, self._clip_ratio, True
                )
                total_loss = policy_loss + wv * value_loss - we * entropy_loss
                self._optimizer.zero_grad(set_to_none=False)
                total_loss.backward()
                if self._optimizer_capturable:
                    # capturable Adam (on-device step counters): grad clip +
                    # update replay inside the graph
                    self._optimizer.step()
                return {
                    'total_loss': total_loss.detach(), 'policy_loss': policy_loss.detach(),
                    'value_loss': value_loss.detach(), 'entropy_loss': entropy_loss.detach(),
                    'approx_kl': approx_kl, 'clipfrac': clipfrac,
                    'adv_max': adv.max().detach(), 'adv_mean': adv.mean().detach(),
                    'value_mean': output['value'].mean().detach(), 'value_max': output['value'].max().detach(),
                }

            self._graphed_step = GraphedStep(step_fn)
        out = self._graphed_step(batch)
        if not self._optimizer_capturable:
            self._optimizer.step()
        return out

    def _forward_learn(self, data: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        data = default_preprocess_learn(data, ignore_done=self._cfg.learn.ignore_done, use_nstep=False)
        if self._cuda:
            data = to_device(data, self._device)
        if isinstance(data['obs'], torch.Tensor):
            data['obs'] = data['obs'].float()
        if 'next_obs' in data and isinstance(data['next_obs'], torch.Tensor):
            data['next_obs'] = data['next_obs'].float()
        if self._channels_last and isinstance(data['obs'], torch.Tensor) and data['obs'].dim() == 4:
            data['obs'] = data['obs'].to(memory_format=torch.channels_last)
            if 'next_obs' in data and isinstance(data['next_obs'], torch.Tensor):
                data['next_obs'] = data['next_obs'].to(memory_format=torch.channels_last)
        self._learn_model.train()
        return_infos = []
        _both_cache = None
        for epoch in range(self._cfg.learn.epoch_per_collect):
            if self._recompute_adv:
                with torch.no_grad():
                    if isinstance(data['obs'], torch.Tensor):
                        # one batched critic pass over [obs; next_obs], chunked
                        # at the learn minibatch size: those conv shapes are
                        # already tuned (rocprof: batch-3200 critic passes fell
                        # to MIOpen's naive fp64-accum convs at ~17 ms each).
                        # Under cuda_graph the whole chunked pass is captured
                        # once and replayed per epoch (~200 eager launches ->
                        # one hipGraphLaunch).
                        if _both_cache is None:
                            _both_cache = torch.cat([data['obs'], data['next_obs']], dim=0)
                        both = _both_cache
                        # chunk-size sweep on MI355X (320/640/1600/2400/3200
                        # -> 224.7/210.7/205.3/210.6/202.8 ms/step): 3200 wins
                        # once the captured values graph + find-mode warmup
                        # keep MIOpen on tuned solvers at that batch
                        import os as _os
                        _chunk = int(_os.environ.get('DING_PPO_VALUE_CHUNK', 0)) or \
                            max(int(self._cfg.learn.batch_size), 3200)
                        if self._cuda_graph and both.is_cuda and both.dtype == torch.float32:
                            values = self._graphed_values(both, _chunk, fresh=(epoch == 0))
                        else:
                            chunks = torch.split(both, _chunk, dim=0)
                            with self._amp_ctx(both):
                                values = torch.cat(
                                    [
                                        self._learn_model.forward(c, mode='compute_critic')['value']
                                        for c in chunks
                                    ],
                                    dim=0
                                ).float()
                        value, next_value = values.chunk(2, dim=0)
                    else:
                        value = self._learn_model.forward(data['obs'], mode='compute_critic')['value']
                        next_value = self._learn_model.forward(data['next_obs'], mode='compute_critic')['value']
                    if self._value_norm:
                        value *= float(self._running_mean_std.std[0])
                        next_value *= float(self._running_mean_std.std[0])
                    traj_flag = data.get('traj_flag')
                    compute_adv_data = gae_data(value, next_value, data['reward'], data['done'], traj_flag)
                    data['adv'] = gae(compute_adv_data, self._gamma, self._gae_lambda)
                    unnormalized_returns = value + data['adv']
                    if self._value_norm:
                        data['value'] = value / float(self._running_mean_std.std[0])
                        data['return'] = unnormalized_returns / float(self._running_mean_std.std[0])
                        self._running_mean_std.update(unnormalized_returns.cpu().numpy().reshape(-1, 1))
                    else:
                        data['value'] = value
                        data['return'] = unnormalized_returns
            else:
                if self._value_norm:
                    unnormalized_return = data['adv'] + data['value'] * float(self._running_mean_std.std[0])
                    data['return'] = unnormalized_return / float(self._running_mean_std.std[0])
                    self._running_mean_std.update(unnormalized_return.cpu().numpy().reshape(-1, 1))
                else:
                    data['return'] = data['adv'] + data['value']

            graph_infos = []
            for batch in split_data_generator(data, self._cfg.learn.batch_size, shuffle=True):
                if (
                    self._cuda_graph and self._action_space == 'discrete'
                    and isinstance(batch['obs'], torch.Tensor) and batch['obs'].is_cuda
                    and batch.get('weight') is None and batch['obs'].dtype == torch.float32
                ):
                    out = self._graphed_minibatch({
                        k: batch[k] for k in ('obs', 'logit', 'action', 'value', 'adv', 'return')
                    })
                    # static outputs: clone (async) now, convert to floats once
                    # at the end of the epoch — avoids a device sync per minibatch
                    # ONE packed stack per minibatch instead of a clone per
                    # stat (rocprof: copyBuffer was ~2.7k launches/step)
                    if not hasattr(self, '_graph_info_keys') or self._graph_info_keys is None:
                        self._graph_info_keys = list(out.keys())
                    graph_infos.append(torch.stack([out[k] for k in self._graph_info_keys]))
                    continue
                with self._amp_ctx(batch['adv']):
                    output = self._learn_model.forward(batch['obs'], mode='compute_actor_critic')
                if self._bf16:  # loss math stays fp32
                    output = {
                        k: (v.float() if isinstance(v, torch.Tensor) else
                            {kk: vv.float() for kk, vv in v.items()} if isinstance(v, dict) else v)
                        for k, v in output.items()
                    }
                adv = batch['adv']
                if self._adv_norm:
                    adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                if self._action_space == 'continuous':
                    ppo_batch = ppo_data(
                        output['logit'], batch['logit'], batch['action'], output['value'], batch['value'], adv,
                        batch['return'], batch.get('weight')
                    )
                    ppo_loss, ppo_info = ppo_error_continuous(ppo_batch, self._clip_ratio)
                elif self._action_space == 'discrete':
                    ppo_batch = ppo_data(
                        output['logit'], batch['logit'], batch['action'], output['value'], batch['value'], adv,
                        batch['return'], batch.get('weight')
                    )
                    ppo_loss, ppo_info = ppo_error(ppo_batch, self._clip_ratio)
                else:  # hybrid
                    # action-type part
                    type_batch = ppo_policy_data(
                        output['logit']['action_type'], batch['logit']['action_type'], batch['action']['action_type'],
                        adv, batch.get('weight')
                    )
                    type_loss, type_info = ppo_policy_error(type_batch, self._clip_ratio)
                    from ding.rl_utils.ppo import ppo_policy_data_continuous, ppo_value_data, ppo_value_error
                    args_batch = ppo_policy_data_continuous(
                        output['logit']['action_args'], batch['logit']['action_args'],
                        batch['action']['action_args'], adv, batch.get('weight')
                    )
                    args_loss, args_info = ppo_policy_error_continuous(args_batch, self._clip_ratio)
                    value_loss = ppo_value_error(
                        ppo_value_data(output['value'], batch['value'], batch['return'], batch.get('weight')),
                        self._clip_ratio
                    )
                    from ding.rl_utils.ppo import ppo_loss as ppo_loss_tuple, ppo_info as ppo_info_tuple
                    ppo_loss = ppo_loss_tuple(
                        type_loss.policy_loss + args_loss.policy_loss, value_loss,
                        type_loss.entropy_loss + args_loss.entropy_loss, torch.zeros(())
                    )
                    ppo_info = ppo_info_tuple(
                        max(type_info.approx_kl, args_info.approx_kl), max(type_info.clipfrac, args_info.clipfrac)
                    )
                wv, we = self._value_weight, self._entropy_weight
                total_loss = ppo_loss.policy_loss + wv * ppo_loss.value_loss - we * ppo_loss.entropy_loss
                self._optimizer.zero_grad()
                total_loss.backward()
                if self._cfg.multi_gpu:
                    self.sync_gradients(self._model)
                self._optimizer.step()

                return_info = {
                    'cur_lr': self._optimizer.defaults['lr'],
                    'total_loss': total_loss.item(),
                    'policy_loss': ppo_loss.policy_loss.item(),
                    'value_loss': ppo_loss.value_loss.item(),
                    'entropy_loss': ppo_loss.entropy_loss.item(),
                    'adv_max': adv.max().item(),
                    'adv_mean': adv.mean().item(),
                    'value_mean': output['value'].mean().item(),
                    'value_max': output['value'].max().item(),
                    'approx_kl': ppo_info.approx_kl,
                    'clipfrac': ppo_info.clipfrac,
                }
                return_infos.append(return_info)
            if graph_infos:
                # one host sync for the whole epoch's graphed minibatches
                lr = self._optimizer.defaults['lr']
                packed = torch.stack(graph_infos).cpu()  # [n_minibatch, n_keys]
                keys = self._graph_info_keys
                for row in packed:
                    info = {k: float(v) for k, v in zip(keys, row)}
                    info['cur_lr'] = lr
                    return_infos.append(info)
        return return_infos

    def _monitor_vars_learn(self) -> List[str]:
        variables = [
            'cur_lr', 'total_loss', 'policy_loss', 'value_loss', 'entropy_loss', 'adv_max', 'adv_mean',
            'approx_kl', 'clipfrac', 'value_max', 'value_mean',
        ]
        return variables

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._action_space = self._cfg.action_space
        if self._action_space == 'continuous':
            self._collect_model = model_wrap(self._model, wrapper_name='reparam_sample')
        elif self._action_space == 'discrete':
            self._collect_model = model_wrap(self._model, wrapper_name='multinomial_sample')
        else:
            self._collect_model = model_wrap(self._model, wrapper_name='hybrid_reparam_multinomial_sample')
        self._collect_model.reset()
        self._gamma = self._cfg.collect.discount_factor
        self._gae_lambda = self._cfg.collect.gae_lambda
        self._recompute_adv = self._cfg.recompute_adv

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        data_id = list(data.keys())
        data = default_collate(list(data.values()))
        if self._cuda:
            data = to_device(data, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(data, mode='compute_actor_critic')
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
        data = transitions
        data = get_gae_with_default_last_value_wrapper(
            data, done=data[-1]['done'], gamma=self._gamma, gae_lambda=self._gae_lambda, cuda=False
        )
        return get_train_sample(data, self._unroll_len)

    def _init_eval(self) -> None:
        self._action_space = self._cfg.action_space
        if self._action_space == 'continuous':
            self._eval_model = model_wrap(self._model, wrapper_name='deterministic_sample')
        elif self._action_space == 'discrete':
            self._eval_model = model_wrap(self._model, wrapper_name='argmax_sample')
        else:
            self._eval_model = model_wrap(self._model, wrapper_name='hybrid_deterministic_argmax_sample')
        self._eval_model.reset()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        data = default_collate(list(data.values()))
        if self._cuda:
            data = to_device(data, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            output = self._eval_model.forward(data, mode='compute_actor')
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}


def get_gae_with_default_last_value_wrapper(data, done, gamma, gae_lambda, cuda):
    from collections import deque
    from ding.rl_utils import Adder
    return Adder.get_gae_with_default_last_value(deque(data), done, gamma, gae_lambda, cuda)


@POLICY_REGISTRY.register('ppo_pg')
class PPOPGPolicy(Policy):
    """PPO policy-gradient-only variant (no value function; MC returns)."""

    config = dict(
        type='ppo_pg',
        cuda=False,
        on_policy=True,
        action_space='discrete',
        model=dict(),
        learn=dict(
            epoch_per_collect=10,
            batch_size=64,
            learning_rate=3e-4,
            entropy_weight=0.01,
            clip_ratio=0.2,
            grad_clip_type='clip_norm',
            grad_clip_value=0.5,
            ignore_done=False,
        ),
        collect=dict(
            unroll_len=1,
            discount_factor=0.99,
        ),
        eval=dict(),
    )

    def default_model(self) -> tuple:
        return 'pg', ['ding.model.template.pg']

    def _init_learn(self) -> None:
        self._optimizer = Adam(
            self._model.parameters(), lr=self._cfg.learn.learning_rate,
            grad_clip_type=self._cfg.learn.grad_clip_type, clip_value=self._cfg.learn.grad_clip_value
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._clip_ratio = self._cfg.learn.clip_ratio
        self._action_space = self._cfg.action_space
        self._learn_model.reset()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        data = default_preprocess_learn(data)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        return_infos = []
        for epoch in range(self._cfg.learn.epoch_per_collect):
            for batch in split_data_generator(data, self._cfg.learn.batch_size, shuffle=True):
                output = self._learn_model.forward(batch['obs'])
                adv = batch['return']
                adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                if self._action_space == 'discrete':
                    pol_data = ppo_policy_data(output['logit'], batch['logit'], batch['action'], adv,
                                               batch.get('weight'))
                    loss_t, info = ppo_policy_error(pol_data, self._clip_ratio)
                else:
                    from ding.rl_utils.ppo import ppo_policy_data_continuous
                    pol_data = ppo_policy_data_continuous(output['logit'], batch['logit'], batch['action'], adv,
                                                          batch.get('weight'))
                    loss_t, info = ppo_policy_error_continuous(pol_data, self._clip_ratio)
                total_loss = loss_t.policy_loss - self._entropy_weight * loss_t.entropy_loss
                self._optimizer.zero_grad()
                total_loss.backward()
                if self._cfg.multi_gpu:
                    self.sync_gradients(self._model)
                self._optimizer.step()
                return_infos.append({
                    'cur_lr': self._optimizer.defaults['lr'],
                    'total_loss': total_loss.item(),
                    'policy_loss': loss_t.policy_loss.item(),
                    'entropy_loss': loss_t.entropy_loss.item(),
                    'approx_kl': info.approx_kl,
                    'clipfrac': info.clipfrac,
                })
        return return_infos

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'policy_loss', 'entropy_loss', 'approx_kl', 'clipfrac']

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
        data = default_collate(list(data.values()))
        if self._cuda:
            data = to_device(data, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(data)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs, policy_output, timestep) -> Dict[str, Any]:
        return {
            'obs': obs,
            'action': policy_output['action'],
            'logit': policy_output['logit'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _get_train_sample(self, data: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        # Monte-Carlo returns over the episode
        R = 0.0
        for i in reversed(range(len(data))):
            R = self._gamma * R + data[i]['reward'].item()
            data[i]['return'] = torch.tensor([R])
        return get_train_sample(data, self._unroll_len)

    def _init_eval(self) -> None:
        if self._cfg.action_space == 'discrete':
            self._eval_model = model_wrap(self._model, wrapper_name='argmax_sample')
        else:
            self._eval_model = model_wrap(self._model, wrapper_name='deterministic_sample')
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


@POLICY_REGISTRY.register('ppo_offpolicy')
class PPOOffPolicy(PPOPolicy):
    """Off-policy PPO: GAE computed at collect time, samples pushed to a
    replay buffer, nstep-return option."""

    config = dict(
        type='ppo_offpolicy',
        on_policy=False,
        recompute_adv=False,
        learn=dict(
            update_per_collect=5,
            batch_size=64,
        ),
        other=dict(replay_buffer=dict(replay_buffer_size=10000, )),
    )

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(data, ignore_done=self._cfg.learn.ignore_done, use_nstep=False)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        data['obs'] = data['obs'].float()
        data['return'] = data['adv'] + data['value']
        output = self._learn_model.forward(data['obs'], mode='compute_actor_critic')
        adv = data['adv']
        if self._adv_norm:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        ppo_batch = ppo_data(
            output['logit'], data['logit'], data['action'], output['value'], data['value'], adv, data['return'],
            data.get('weight')
        )
        if self._action_space == 'continuous':
            ppo_loss, ppo_info = ppo_error_continuous(ppo_batch, self._clip_ratio)
        else:
            ppo_loss, ppo_info = ppo_error(ppo_batch, self._clip_ratio)
        total_loss = ppo_loss.policy_loss + self._value_weight * ppo_loss.value_loss \
            - self._entropy_weight * ppo_loss.entropy_loss
        self._optimizer.zero_grad()
        total_loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': total_loss.item(),
            'policy_loss': ppo_loss.policy_loss.item(),
            'value_loss': ppo_loss.value_loss.item(),
            'entropy_loss': ppo_loss.entropy_loss.item(),
            'approx_kl': ppo_info.approx_kl,
            'clipfrac': ppo_info.clipfrac,
        }

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        data = get_gae_with_default_last_value_wrapper(
            transitions, done=transitions[-1]['done'], gamma=self._gamma, gae_lambda=self._gae_lambda, cuda=False
        )
        return get_train_sample(data, self._unroll_len)


@POLICY_REGISTRY.register('ppo_stdim')
class PPOSTDIMPolicy(PPOPolicy):
    """PPO + ST-DIM auxiliary contrastive representation loss over the
    actor-critic encoder.

    Parity: reference ding/policy/ppo.py PPOSTDIMPolicy ('ppo_stdim':1591).
    """

    config = dict(
        type='ppo_stdim',
        aux_loss_weight=0.001,
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        from ding.torch_utils.loss import ContrastiveLoss
        x_size, y_size = self._get_encoding_size()
        self._aux_model = ContrastiveLoss(x_size, y_size)
        if self._cuda:
            self._aux_model.cuda()
        self._aux_optimizer = Adam(self._aux_model.parameters(), lr=self._cfg.learn.learning_rate)
        self._aux_loss_weight = self._cfg.aux_loss_weight

    def _encoder(self):
        # VAC: shared encoder or actor-side encoder
        enc = getattr(self._model, 'encoder', None)
        if enc is None:
            enc = self._model.actor_encoder
        return enc

    def _get_encoding_size(self):
        obs = self._cfg.model.obs_shape
        test = torch.randn(1, obs) if isinstance(obs, int) else torch.randn(1, *obs)
        if self._cuda:
            test = test.cuda()
        with torch.no_grad():
            x = self._encoder()(test)
        return x.shape[1], x.shape[1]

    def _forward_learn(self, data) -> List[Dict[str, Any]]:
        collated = default_preprocess_learn(data, ignore_done=self._cfg.learn.ignore_done, use_nstep=False)
        if self._cuda:
            collated = to_device(collated, self._device)
        with torch.no_grad():
            x = self._encoder()(collated['obs'].float())
            y = self._encoder()(collated['next_obs'].float())
        aux_loss = self._aux_model(x, y) * self._aux_loss_weight
        self._aux_optimizer.zero_grad()
        aux_loss.backward()
        self._aux_optimizer.step()
        out = super()._forward_learn(data)
        for info in out if isinstance(out, list) else [out]:
            info['aux_loss'] = aux_loss.item()
        return out

    def _monitor_vars_learn(self) -> List[str]:
        return super()._monitor_vars_learn() + ['aux_loss']
