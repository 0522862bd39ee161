"""PPOF: the simplified, flat-config PPO used by the high-level bonus API.

Parity: reference ding/policy/ppof.py ('ppof':18). Differences by design:
batches are plain dicts of tensors (no treetensor), the model is any
VAC-style module exposing compute_actor/compute_critic/compute_actor_critic,
and the discrete learn path reuses the fused HIP PPO loss via ppo_error.
"""
import copy
from collections import namedtuple
from typing import Any, Callable, Dict, List, Optional

import torch
from torch.optim import AdamW

from ding.rl_utils import (
    ArgmaxSampler, MultinomialSampler, MuSampler, ReparameterizationSampler, gae, gae_data, inv_symlog, ppo_data,
    ppo_error, ppo_error_continuous, symlog, value_inv_transform, value_transform,
)
from ding.utils import POLICY_REGISTRY, EasyDict, RunningMeanStd


@POLICY_REGISTRY.register('ppof')
class PPOFPolicy:

    config = dict(
        type='ppof',
        on_policy=True,
        cuda=True,
        action_space='discrete',
        discount_factor=0.99,
        gae_lambda=0.95,
        epoch_per_collect=10,
        batch_size=64,
        learning_rate=3e-4,
        lr_scheduler=None,
        weight_decay=0,
        value_weight=0.5,
        entropy_weight=0.01,
        clip_ratio=0.2,
        adv_norm=True,
        value_norm='baseline',
        ppo_param_init=True,
        grad_norm=0.5,
        n_sample=128,
        unroll_len=1,
        deterministic_eval=True,
        model=dict(),
    )
    mode = ['learn', 'collect', 'eval']

    @classmethod
    def default_config(cls) -> EasyDict:
        cfg = EasyDict(copy.deepcopy(cls.config))
        cfg.cfg_type = cls.__name__ + 'Dict'
        return cfg

    @classmethod
    def default_model(cls) -> Callable:
        from ding.model.template.vac import VAC
        return VAC

    def __init__(self, cfg: EasyDict, model: torch.nn.Module, enable_mode: Optional[List[str]] = None):
        self._cfg = cfg
        self._model = model
        self._device = 'cuda' if (cfg.cuda and torch.cuda.is_available()) else 'cpu'
        self._model.to(self._device)
        assert cfg.action_space in ('discrete', 'continuous')
        self._action_space = cfg.action_space
        if cfg.ppo_param_init:
            self._model_param_init()
        self.enable_mode = enable_mode or self.mode
        if 'learn' in self.enable_mode:
            self._optimizer = AdamW(
                self._model.parameters(), lr=cfg.learning_rate, weight_decay=cfg.weight_decay
            )
            self._lr_scheduler = None
            if cfg.lr_scheduler is not None:
                milestone, factor = cfg.lr_scheduler
                self._lr_scheduler = torch.optim.lr_scheduler.StepLR(
                    self._optimizer, step_size=milestone, gamma=factor
                )
            if cfg.value_norm == 'baseline':
                self._running_mean_std = RunningMeanStd(epsilon=1e-4)
        if 'collect' in self.enable_mode:
            if self._action_space == 'discrete':
                self._collect_sampler = MultinomialSampler()
            else:
                self._collect_sampler = ReparameterizationSampler()
        if 'eval' in self.enable_mode:
            if self._action_space == 'discrete':
                self._eval_sampler = ArgmaxSampler() if cfg.deterministic_eval else MultinomialSampler()
            else:
                self._eval_sampler = MuSampler() if cfg.deterministic_eval else ReparameterizationSampler()

    def _model_param_init(self):
        import numpy as np
        for m in self._model.modules():
            if isinstance(m, torch.nn.Linear):
                torch.nn.init.orthogonal_(m.weight, gain=np.sqrt(2))
                torch.nn.init.zeros_(m.bias)

    # ----------------------------------------------------------------- learn
    def forward(self, data: Dict[str, torch.Tensor]) -> List[Dict[str, Any]]:
        """data: dict of stacked tensors {obs, next_obs, action, logit,
        reward, done[, traj_flag]}; runs epoch_per_collect epochs of
        recompute-adv PPO and returns per-minibatch logs."""
        return_infos = []
        self._model.train()
        cfg = self._cfg
        n = (data['obs'].shape[0] // cfg.batch_size) * cfg.batch_size
        data = {k: (v[:n] if isinstance(v, torch.Tensor) else v) for k, v in data.items()}
        data = {k: (v.to(self._device) if isinstance(v, torch.Tensor) else v) for k, v in data.items()}

        for epoch in range(cfg.epoch_per_collect):
            with torch.no_grad():
                value = self._model.forward(data['obs'], mode='compute_critic')['value']
                next_value = self._model.forward(data['next_obs'], mode='compute_critic')['value']
                reward = data['reward']
                if cfg.value_norm == 'value_rescale':
                    value, next_value = value_inv_transform(value), value_inv_transform(next_value)
                elif cfg.value_norm == 'symlog':
                    value, next_value = inv_symlog(value), inv_symlog(next_value)
                elif cfg.value_norm == 'baseline':
                    std = float(self._running_mean_std.std)
                    value, next_value = value * std, next_value * std
                adv_data = gae_data(value, next_value, reward, data['done'].float(), data.get('traj_flag'))
                data['adv'] = gae(adv_data, cfg.discount_factor, cfg.gae_lambda)
                unnormalized_returns = value + data['adv']
                if cfg.value_norm == 'value_rescale':
                    value = value_transform(value)
                    unnormalized_returns = value_transform(unnormalized_returns)
                elif cfg.value_norm == 'symlog':
                    value = symlog(value)
                    unnormalized_returns = symlog(unnormalized_returns)
                elif cfg.value_norm == 'baseline':
                    value = value / std
                    unnormalized_returns = unnormalized_returns / std
                    self._running_mean_std.update((unnormalized_returns * std).cpu().numpy())
                data['value'] = value
                data['return_'] = unnormalized_returns

            B = data['obs'].shape[0]
            perm = torch.randperm(B, device=data['obs'].device)
            for start in range(0, B, cfg.batch_size):
                idx = perm[start:start + cfg.batch_size]
                batch = {k: (v[idx] if isinstance(v, torch.Tensor) else v) for k, v in data.items()}
                output = self._model.forward(batch['obs'], mode='compute_actor_critic')
                adv = batch['adv']
                if cfg.adv_norm:
                    adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                ppo_batch = ppo_data(
                    output['logit'], batch['logit'], batch['action'], output['value'], batch['value'], adv,
                    batch['return_'], None
                )
                if self._action_space == 'discrete':
                    ppo_loss, ppo_info = ppo_error(ppo_batch, cfg.clip_ratio)
                else:
                    ppo_loss, ppo_info = ppo_error_continuous(ppo_batch, cfg.clip_ratio)
                wv, we = cfg.value_weight, cfg.entropy_weight
                total_loss = ppo_loss.policy_loss + wv * ppo_loss.value_loss - we * ppo_loss.entropy_loss
                self._optimizer.zero_grad()
                total_loss.backward()
                torch.nn.utils.clip_grad_norm_(self._model.parameters(), cfg.grad_norm)
                self._optimizer.step()
                return_infos.append({
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
                })
        if self._lr_scheduler is not None:
            self._lr_scheduler.step()
        return return_infos

    # --------------------------------------------------------------- collect
    def collect(self, data: torch.Tensor) -> Dict[str, torch.Tensor]:
        self._model.eval()
        with torch.no_grad():
            output = self._model.forward(data.to(self._device), mode='compute_actor_critic')
            action = self._collect_sampler(output['logit'])
        return {'action': action, 'logit': output['logit'], 'value': output['value']}

    def process_transition(self, obs, inference_output: dict, timestep: namedtuple) -> Dict[str, Any]:
        return {
            'obs': obs,
            'next_obs': timestep.obs,
            'action': inference_output['action'],
            'logit': inference_output['logit'],
            'value': inference_output['value'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    # ------------------------------------------------------------------ eval
    def eval(self, data: torch.Tensor) -> Dict[str, torch.Tensor]:
        self._model.eval()
        with torch.no_grad():
            logit = self._model.forward(data.to(self._device), mode='compute_actor')['logit']
            action = self._eval_sampler(logit)
        return {'action': action, 'logit': logit}

    # ------------------------------------------------------------------ misc
    def state_dict(self) -> Dict[str, Any]:
        sd = {'model': self._model.state_dict()}
        if 'learn' in self.enable_mode:
            sd['optimizer'] = self._optimizer.state_dict()
        return sd

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        self._model.load_state_dict(state_dict['model'])
        if 'learn' in self.enable_mode and 'optimizer' in state_dict:
            self._optimizer.load_state_dict(state_dict['optimizer'])

    def monitor_vars(self) -> List[str]:
        return [
            'cur_lr', 'total_loss', 'policy_loss', 'value_loss', 'entropy_loss', 'adv_max', 'adv_mean',
            'value_mean', 'value_max', 'approx_kl', 'clipfrac'
        ]

    def reset(self, env_id_list: Optional[List[int]] = None) -> None:
        pass
