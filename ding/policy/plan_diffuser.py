"""Plan Diffuser (PD): offline trajectory-diffusion planning policy.

Parity: reference ding/policy/plan_diffuser.py ('pd':19). Learn consumes
offline batches {'trajectories', 'condition_id', 'condition_val',
'returns'?}; eval denoises a plan conditioned on the current (and optionally
goal) observation and extracts the first action (value-guided) or follows
waypoints (goal-conditioned maze mode).
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List, Optional, Tuple, Union

import numpy as np
import torch

from ding.model import model_wrap
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn


class _IdentityNormalizer:

    def normalize(self, x, key=None):
        return x

    def unnormalize(self, x, key=None):
        return x


@POLICY_REGISTRY.register('pd')
class PDPolicy(Policy):

    config = dict(
        type='pd',
        cuda=False,
        on_policy=False,
        priority=False,
        priority_IS_weight=False,
        model=dict(
            diffuser_model='GaussianDiffusion',
            diffuser_model_cfg=dict(),
            value_model=None,
            value_model_cfg=None,
        ),
        learn=dict(
            batch_size=64,
            learning_rate=2e-4,
            gradient_accumulate_every=2,
            plan_batch_size=64,
            update_target_freq=10,
            step_start_update_target=2000,
            target_weight=0.995,
            value_step=200e3,
            include_returns=True,
            discount_factor=0.99,
            ignore_done=False,
            learner=dict(hook=dict(log_show_after_iter=1000)),
        ),
        collect=dict(unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=5000)),
        other=dict(replay_buffer=dict(replay_buffer_size=1000000)),
    )

    def default_model(self) -> Tuple[str, List[str]]:
        return 'pd', ['ding.model.template.diffusion']

    def _init_learn(self) -> None:
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self.action_dim = self._cfg.model.diffuser_model_cfg.action_dim
        self.obs_dim = self._cfg.model.diffuser_model_cfg.obs_dim
        self.n_timesteps = self._cfg.model.diffuser_model_cfg.n_timesteps
        self.gradient_accumulate_every = self._cfg.learn.gradient_accumulate_every
        self.plan_batch_size = self._cfg.learn.plan_batch_size
        self.gradient_steps = 1
        self.update_target_freq = self._cfg.learn.update_target_freq
        self.step_start_update_target = self._cfg.learn.step_start_update_target
        self.target_weight = self._cfg.learn.target_weight
        self.value_step = self._cfg.learn.value_step
        self.use_target = False
        self.horizon = self._cfg.model.diffuser_model_cfg.horizon
        self._plan_optimizer = Adam(self._model.diffuser.model.parameters(), lr=self._cfg.learn.learning_rate)
        if self._model.value:
            self._value_optimizer = Adam(self._model.value.model.parameters(), lr=self._cfg.learn.learning_rate)
        self._gamma = self._cfg.learn.discount_factor
        self._target_model = copy.deepcopy(self._model)
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.reset()
        self._forward_learn_cnt = 0
        self.normalizer = _IdentityNormalizer()

    def init_data_normalizer(self, normalizer=None) -> None:
        if normalizer is not None:
            self.normalizer = normalizer

    def _forward_learn(self, data: Union[dict, List[dict]]) -> Dict[str, Any]:
        loss_dict = {}
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight,
            ignore_done=self._cfg.learn.ignore_done, use_nstep=False
        )
        conds = {}
        vals = data['condition_val']
        ids = data['condition_id']
        if isinstance(ids, torch.Tensor) and ids.dim() == 1:
            # one condition slot per sample (collated to [B]): e.g. {0: s_0}
            conds[int(ids[0])] = vals
        else:
            # multiple slots: ids[i] is the i-th slot's [B] ids, vals[i] its values
            for i in range(len(ids)):
                key = ids[i][0].item() if isinstance(ids[i], torch.Tensor) else int(ids[i][0])
                conds[key] = vals[i]
            if len(ids) > 1:
                self.use_target = True
        data['conditions'] = conds
        if 'returns' in data.keys() and data['returns'].dim() == 1:
            data['returns'] = data['returns'].unsqueeze(-1)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        x = data['trajectories']
        batch_size = len(x)
        t = torch.randint(0, self.n_timesteps, (batch_size, ), device=x.device).long()
        cond = data['conditions']
        loss_dict['diffuse_loss'], a0 = self._model.diffuser_loss(x, cond, t)
        loss_dict['a0_loss'] = float(a0.detach()) if isinstance(a0, torch.Tensor) else float(a0)
        loss_dict['diffuse_loss'] = loss_dict['diffuse_loss'] / self.gradient_accumulate_every
        loss_dict['diffuse_loss'].backward()
        if self._forward_learn_cnt < self.value_step and self._model.value:
            target = data['returns']
            vloss, logs = self._model.value_loss(x, cond, target, t)
            vloss = vloss / self.gradient_accumulate_every
            vloss.backward()
            loss_dict['value_loss'] = float(vloss)
            loss_dict.update(logs)
        if self.gradient_steps >= self.gradient_accumulate_every:
            self._plan_optimizer.step()
            self._plan_optimizer.zero_grad()
            if self._forward_learn_cnt < self.value_step and self._model.value:
                self._value_optimizer.step()
                self._value_optimizer.zero_grad()
            self.gradient_steps = 1
        else:
            self.gradient_steps += 1
        self._forward_learn_cnt += 1
        if self._forward_learn_cnt % self.update_target_freq == 0:
            if self._forward_learn_cnt < self.step_start_update_target:
                self._target_model.load_state_dict(self._model.state_dict())
            else:
                self.update_model_average(self._target_model, self._learn_model)
        loss_dict['diffuse_loss'] = float(loss_dict['diffuse_loss'])
        loss_dict['mean_traj'] = float(x.mean())
        return loss_dict

    def update_model_average(self, ma_model, current_model) -> None:
        """EMA of learner params into the target planner."""
        for cur, ma in zip(current_model.parameters(), ma_model.parameters()):
            if ma.data is None:
                ma.data = cur.data
            else:
                ma.data = self.target_weight * ma.data + (1 - self.target_weight) * cur.data

    def _monitor_vars_learn(self) -> List[str]:
        return ['diffuse_loss', 'a0_loss', 'value_loss', 'mean_pred', 'max_pred', 'min_pred', 'mean_traj']

    def _state_dict_learn(self) -> Dict[str, Any]:
        return {
            'model': self._learn_model.state_dict(),
            'target_model': self._target_model.state_dict(),
            'plan_optimizer': self._plan_optimizer.state_dict(),
        }

    def _load_state_dict_learn(self, state_dict: Dict[str, Any]) -> None:
        self._learn_model.load_state_dict(state_dict['model'])
        self._target_model.load_state_dict(state_dict['target_model'])
        self._plan_optimizer.load_state_dict(state_dict['plan_optimizer'])

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='base')
        self._eval_model.reset()
        self._plan_seq = []
        self._eval_t = []
        if not hasattr(self, 'normalizer'):
            self.normalizer = _IdentityNormalizer()

    def _forward_eval(self, data: dict) -> Dict[str, Any]:
        data_id = list(data.keys())
        data = default_collate(list(data.values()))
        self._eval_model.eval()
        with torch.no_grad():
            if self.use_target:
                cur_obs = torch.as_tensor(self.normalizer.normalize(data[:, :self.obs_dim], 'observations'))
                target_obs = torch.as_tensor(self.normalizer.normalize(data[:, self.obs_dim:], 'observations'))
                if self._cuda:
                    cur_obs, target_obs = to_device(cur_obs, self._device), to_device(target_obs, self._device)
                conditions = {0: cur_obs.float(), self.horizon - 1: target_obs.float()}
                if self._plan_seq == [] or 0 in self._eval_t:
                    plan = self._eval_model.get_eval(conditions, self.plan_batch_size)
                    plan = to_device(plan, 'cpu').numpy()
                    if self._plan_seq == []:
                        self._plan_seq = plan
                        self._eval_t = [0] * len(data_id)
                    else:
                        for i in data_id:
                            if self._eval_t[i] == 0:
                                self._plan_seq[i] = plan[i]
                action = []
                for i in data_id:
                    if self._eval_t[i] < len(self._plan_seq[i]) - 1:
                        next_wp = self._plan_seq[i][self._eval_t[i] + 1]
                    else:
                        next_wp = self._plan_seq[i][-1].copy()
                        next_wp[2:] = 0
                    cur = to_device(cur_obs[i], 'cpu').numpy()
                    action.append(next_wp[:2] - cur[:2] + (next_wp[2:] - cur[2:]))
                    self._eval_t[i] += 1
                action = torch.as_tensor(np.stack(action))
            else:
                obs = torch.as_tensor(self.normalizer.normalize(data, 'observations')).float()
                if self._cuda:
                    obs = to_device(obs, self._device)
                conditions = {0: obs}
                action = self._eval_model.get_eval(conditions, self.plan_batch_size)
                if self._cuda:
                    action = to_device(action, 'cpu')
                action = torch.as_tensor(self.normalizer.unnormalize(action, 'actions'))
        output = default_decollate({'action': action})
        return {i: d for i, d in zip(data_id, output)}

    def _reset_eval(self, data_id: Optional[List[int]] = None) -> None:
        if self.use_target and data_id:
            for i in data_id:
                if i < len(self._eval_t):
                    self._eval_t[i] = 0

    def _init_collect(self) -> None:
        pass

    def _forward_collect(self, data: dict, **kwargs) -> dict:
        return self._forward_eval(data)

    def _process_transition(self, obs: Any, model_output: dict, timestep: namedtuple) -> dict:
        return {
            'obs': obs, 'action': model_output['action'], 'reward': timestep.reward, 'done': timestep.done,
        }

    def _get_train_sample(self, data: list) -> Union[None, List[Any]]:
        from ding.rl_utils import get_train_sample
        return get_train_sample(data, self._unroll_len if hasattr(self, '_unroll_len') else 1)
