"""Decision Transformer policy (offline sequence modelling + RTG-conditioned
eval).

Parity: reference ding/policy/dt.py ('dt').
"""
from typing import Any, Dict, List

import torch
import torch.nn.functional as F

from ding.model import model_wrap
from ding.torch_utils import to_device
from ding.utils import POLICY_REGISTRY
from .base_policy import Policy


@POLICY_REGISTRY.register('dt')
class DTPolicy(Policy):

    config = dict(
        type='dt',
        cuda=False,
        on_policy=False,
        obs_shape=4,
        action_shape=2,
        rtg_target=200,
        max_eval_ep_len=200,
        rtg_scale=1000,
        context_len=20,
        model=dict(),
        learn=dict(
            batch_size=64,
            learning_rate=1e-4,
            wt_decay=1e-4,
            warmup_steps=10000,
            clip_grad_norm_p=0.25,
        ),
        collect=dict(unroll_len=1, ),
        eval=dict(evaluator=dict(eval_freq=100, )),
    )

    def default_model(self) -> tuple:
        return 'dt', ['ding.model.template.decision_transformer']

    def _init_learn(self) -> None:
        self._optimizer = torch.optim.AdamW(
            self._model.parameters(), lr=self._cfg.learn.learning_rate, weight_decay=self._cfg.learn.wt_decay
        )
        warmup = self._cfg.learn.warmup_steps
        self._scheduler = torch.optim.lr_scheduler.LambdaLR(
            self._optimizer, lambda steps: min((steps + 1) / warmup, 1)
        )
        self._learn_model = self._model
        self._continuous = self._cfg.model.get('continuous', False)
        self._learn_model.train()

    def _forward_learn(self, data: List[tuple]) -> Dict[str, Any]:
        if isinstance(data, list):
            timesteps = torch.stack([d[0] for d in data])
            states = torch.stack([d[1] for d in data])
            actions = torch.stack([d[2] for d in data])
            rtg = torch.stack([d[3] for d in data])
            mask = torch.stack([d[4] for d in data])
        else:
            timesteps, states, actions, rtg, mask = data
        if self._cuda:
            timesteps, states, actions, rtg, mask = to_device(
                (timesteps, states, actions, rtg, mask), self._device
            )
        state_preds, action_preds, return_preds = self._learn_model(timesteps, states, actions, rtg)
        if self._continuous:
            action_target = actions
            loss = F.mse_loss(action_preds, action_target, reduction='none')
            loss = (loss.mean(-1) * mask).sum() / mask.sum().clamp(min=1)
        else:
            action_target = actions.long()
            loss = F.cross_entropy(
                action_preds.reshape(-1, action_preds.shape[-1]), action_target.reshape(-1), reduction='none'
            )
            loss = (loss.reshape(mask.shape) * mask).sum() / mask.sum().clamp(min=1)
        self._optimizer.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self._model.parameters(), self._cfg.learn.clip_grad_norm_p)
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._scheduler.step()
        return {'cur_lr': self._scheduler.get_last_lr()[0], 'total_loss': loss.item()}

    def _init_collect(self) -> None:
        pass

    def _forward_collect(self, data, **kwargs):
        raise NotImplementedError("DT is offline-only; use eval")

    def _process_transition(self, obs, policy_output, timestep):
        raise NotImplementedError

    def _get_train_sample(self, transitions):
        raise NotImplementedError

    def _init_eval(self) -> None:
        """RTG-conditioned autoregressive evaluation state per env."""
        self._eval_model = self._model
        self._context_len = self._cfg.context_len
        self._rtg_target = self._cfg.rtg_target
        self._rtg_scale = self._cfg.rtg_scale
        self._max_len = self._cfg.max_eval_ep_len
        self._continuous = self._cfg.model.get('continuous', False)
        self._eval_state = {}

    def _reset_eval(self, data_id=None):
        if data_id is None:
            self._eval_state = {}
        else:
            for i in data_id:
                self._eval_state.pop(i, None)

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        self._eval_model.eval()
        out = {}
        act_dim = self._cfg.model.act_dim
        state_dim = self._cfg.model.state_dim
        with torch.no_grad():
            for env_id, obs in data.items():
                if env_id not in self._eval_state:
                    self._eval_state[env_id] = {
                        'states': [], 'actions': [], 'rtg': [], 't': 0,
                        'running_rtg': self._rtg_target / self._rtg_scale,
                    }
                st = self._eval_state[env_id]
                st['states'].append(obs.float().reshape(-1))
                st['rtg'].append(torch.tensor([st['running_rtg']]))
                if self._continuous:
                    st['actions'].append(torch.zeros(act_dim))
                else:
                    st['actions'].append(torch.tensor(0))
                C = self._context_len
                states = torch.stack(st['states'][-C:]).unsqueeze(0)
                actions = torch.stack(st['actions'][-C:]).unsqueeze(0)
                rtg = torch.stack(st['rtg'][-C:]).unsqueeze(0)
                T = states.shape[1]
                timesteps = torch.arange(max(0, st['t'] - C + 1), max(0, st['t'] - C + 1) + T).unsqueeze(0)
                if self._cuda:
                    timesteps, states, actions, rtg = to_device((timesteps, states, actions, rtg), self._device)
                _, action_preds, _ = self._eval_model(timesteps, states, actions, rtg)
                pred = action_preds[0, -1]
                if self._continuous:
                    action = pred.cpu()
                else:
                    action = pred.argmax().cpu()
                st['actions'][-1] = action if self._continuous else action
                st['t'] += 1
                out[env_id] = {'action': action}
        return out
