"""HAPPO: heterogeneous-agent PPO with sequential per-agent updates and the
compounding correction factor.

Parity: reference ding/policy/happo.py ('happo').
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap, create_model
from ding.rl_utils import happo_data, happo_error, gae, gae_data
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY, EasyDict, split_data_generator
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn


@POLICY_REGISTRY.register('happo')
class HAPPOPolicy(Policy):
    """One actor-critic (MAVAC) per agent; learn() iterates agents in random
    order, multiplying the surrogate by the ratio factor accumulated from
    already-updated agents."""

    config = dict(
        type='happo',
        cuda=False,
        on_policy=True,
        priority=False,
        action_space='discrete',
        agent_num=2,
        model=dict(),
        learn=dict(
            epoch_per_collect=2,
            batch_size=64,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=True,
            ignore_done=False,
            grad_clip_type='clip_norm',
            grad_clip_value=0.5,
        ),
        collect=dict(unroll_len=1, discount_factor=0.99, gae_lambda=0.95, ),
        eval=dict(),
    )

    def _create_model(self, cfg: EasyDict, model=None):
        """N independent MAVAC models, one per agent."""
        if model is not None:
            return model
        import torch.nn as nn
        agent_num = cfg.agent_num
        models = []
        for _ in range(agent_num):
            m_cfg = EasyDict(copy.deepcopy(cfg.model))
            m_cfg.type = m_cfg.get('type', 'mavac')
            m_cfg.import_names = ['ding.model.template.mavac']
            models.append(create_model(m_cfg))
        return nn.ModuleList(models)

    def default_model(self) -> tuple:
        return 'mavac', ['ding.model.template.mavac']

    def _init_learn(self) -> None:
        self._agent_num = self._cfg.agent_num
        self._optimizers = [
            Adam(m.parameters(), lr=self._cfg.learn.learning_rate, grad_clip_type=self._cfg.learn.grad_clip_type,
                 clip_value=self._cfg.learn.grad_clip_value) for m in self._model
        ]
        self._optimizer = self._optimizers[0]
        self._learn_model = self._model
        self._clip_ratio = self._cfg.learn.clip_ratio
        self._value_weight = self._cfg.learn.value_weight
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._adv_norm = self._cfg.learn.adv_norm
        self._gamma = self._cfg.collect.discount_factor
        self._gae_lambda = self._cfg.collect.gae_lambda

    def _agent_view(self, data: dict, i: int) -> dict:
        """Slice per-agent fields from [B, A, ...] tensors."""
        out = {}
        for k, v in data.items():
            if k == 'obs' and isinstance(v, dict):
                out[k] = {
                    'agent_state': v['agent_state'][:, i],
                    'global_state': v['global_state'],
                    'action_mask': v['action_mask'][:, i] if 'action_mask' in v else None,
                }
            elif isinstance(v, torch.Tensor) and v.dim() >= 2 and v.shape[1] == self._agent_num:
                out[k] = v[:, i]
            else:
                out[k] = v
        return out

    def _forward_learn(self, data: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        data = default_preprocess_learn(data, ignore_done=self._cfg.learn.ignore_done, use_nstep=False)
        if self._cuda:
            data = to_device(data, self._device)
        infos = []
        B = data['action'].shape[0]
        factor = torch.ones(B, 1, device=data['action'].device)
        agent_order = torch.randperm(self._agent_num).tolist()
        for i in agent_order:
            model = self._model[i]
            opt = self._optimizers[i]
            view = self._agent_view(data, i)
            obs_in = view['obs'] if isinstance(view['obs'], dict) else {'agent_state': view['obs'],
                                                                        'global_state': view['obs']}
            # old log-prob before this agent's update (for factor refresh)
            with torch.no_grad():
                old_out = model.forward(obs_in, mode='compute_actor')
                old_dist = torch.distributions.Categorical(logits=old_out['logit'])
                logp_before = old_dist.log_prob(view['action'])
            for epoch in range(self._cfg.learn.epoch_per_collect):
                out = model.forward(obs_in, mode='compute_actor_critic')
                adv = view['adv']
                if self._adv_norm:
                    adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                ret = view['value'] + view['adv']
                hdata = happo_data(
                    out['logit'], view['logit'], view['action'], out['value'], view['value'], adv, ret,
                    view.get('weight'), factor
                )
                loss, info = happo_error(hdata, self._clip_ratio)
                total = loss.policy_loss + self._value_weight * loss.value_loss \
                    - self._entropy_weight * loss.entropy_loss
                opt.zero_grad()
                total.backward()
                if self._cfg.multi_gpu:
                    self.sync_gradients(model)
                opt.step()
                infos.append({
                    'agent': i,
                    'total_loss': total.item(),
                    'policy_loss': loss.policy_loss.item(),
                    'value_loss': loss.value_loss.item(),
                    'entropy_loss': loss.entropy_loss.item(),
                    'approx_kl': info.approx_kl,
                    'clipfrac': info.clipfrac,
                    'cur_lr': opt.defaults['lr'],
                })
            # refresh the factor with this agent's post-update ratio
            with torch.no_grad():
                new_out = model.forward(obs_in, mode='compute_actor')
                new_dist = torch.distributions.Categorical(logits=new_out['logit'])
                logp_after = new_dist.log_prob(view['action'])
                factor = factor * torch.exp(logp_after - logp_before).unsqueeze(-1)
        return infos

    def _monitor_vars_learn(self) -> List[str]:
        return ['total_loss', 'policy_loss', 'value_loss', 'entropy_loss', 'approx_kl', 'clipfrac', 'cur_lr']

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.collect.discount_factor
        self._gae_lambda = self._cfg.collect.gae_lambda

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        outputs = {'logit': [], 'action': [], 'value': []}
        with torch.no_grad():
            for i, model in enumerate(self._model):
                obs_in = {
                    'agent_state': collated['agent_state'][:, i],
                    'global_state': collated['global_state'],
                    'action_mask': collated['action_mask'][:, i] if 'action_mask' in collated else None,
                }
                out = model.forward(obs_in, mode='compute_actor_critic')
                dist = torch.distributions.Categorical(logits=out['logit'])
                outputs['logit'].append(out['logit'])
                outputs['action'].append(dist.sample())
                outputs['value'].append(out['value'])
        merged = {
            'logit': torch.stack(outputs['logit'], dim=1),
            'action': torch.stack(outputs['action'], dim=1),
            'value': torch.stack(outputs['value'], dim=1).mean(dim=1),  # joint value
        }
        if self._cuda:
            merged = to_device(merged, 'cpu')
        out = default_decollate(merged)
        return {i: d for i, d in zip(data_id, out)}

    def _process_transition(self, obs, policy_output, timestep) -> Dict[str, Any]:
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
        from ding.rl_utils import Adder, get_train_sample
        data = Adder.get_gae_with_default_last_value(
            deque(transitions), transitions[-1]['done'], self._gamma, self._gae_lambda, cuda=False
        )
        return get_train_sample(data, self._unroll_len)

    def _init_eval(self) -> None:
        pass

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        actions = []
        with torch.no_grad():
            for i, model in enumerate(self._model):
                obs_in = {
                    'agent_state': collated['agent_state'][:, i],
                    'global_state': collated['global_state'],
                    'action_mask': collated['action_mask'][:, i] if 'action_mask' in collated else None,
                }
                out = model.forward(obs_in, mode='compute_actor')
                actions.append(out['logit'].argmax(dim=-1))
        merged = {'action': torch.stack(actions, dim=1)}
        if self._cuda:
            merged = to_device(merged, 'cpu')
        out = default_decollate(merged)
        return {i: d for i, d in zip(data_id, out)}

    def _state_dict_learn(self) -> Dict[str, Any]:
        return {
            'model': self._model.state_dict(),
            'optimizer': [o.state_dict() for o in self._optimizers],
        }

    def _load_state_dict_learn(self, state_dict: Dict[str, Any]) -> None:
        self._model.load_state_dict(state_dict['model'])
        if 'optimizer' in state_dict:
            for o, sd in zip(self._optimizers, state_dict['optimizer']):
                o.load_state_dict(sd)
