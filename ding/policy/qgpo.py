"""QGPO policy: offline training of a score-based behavior policy, an
in-sample-softmax TwinQ, and a CEP energy guidance; eval samples actions
through the guided diffusion.

Parity: reference ding/policy/qgpo.py ('qgpo':13).
"""
from typing import Any, Dict, List

import torch

from ding.torch_utils import to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate
from .base_policy import Policy


@POLICY_REGISTRY.register('qgpo')
class QGPOPolicy(Policy):

    config = dict(
        type='qgpo',
        cuda=False,
        on_policy=False,
        multi_gpu=False,
        model=dict(qgpo_critic=dict(alpha=3, q_alpha=1)),
        learn=dict(
            learning_rate=1e-4,
            batch_size=4096,
            behavior_policy_stop_training_iter=600000,
            energy_guided_policy_begin_training_iter=600000,
            q_value_stop_training_iter=1100000,
        ),
        eval=dict(guidance_scale=[0.0, 1.0, 2.0], diffusion_steps=15, evaluator=dict(eval_freq=5000)),
    )

    def _init_learn(self) -> None:
        self.cuda = self._cfg.cuda
        self.behavior_model_optimizer = torch.optim.Adam(
            self._model.score_model.parameters(), lr=self._cfg.learn.learning_rate
        )
        self.q_optimizer = torch.optim.Adam(self._model.q.q0.parameters(), lr=3e-4)
        self.qt_optimizer = torch.optim.Adam(self._model.q.qt.parameters(), lr=3e-4)
        self.qt_update_momentum = 0.005
        self.discount = 0.99
        self.behavior_policy_stop_training_iter = self._cfg.learn.behavior_policy_stop_training_iter
        self.energy_guided_policy_begin_training_iter = self._cfg.learn.energy_guided_policy_begin_training_iter
        self.q_value_stop_training_iter = self._cfg.learn.q_value_stop_training_iter

    def _forward_learn(self, data: dict) -> Dict[str, Any]:
        """data keys: s, a, r, s_, d, fake_a (action support for the current
        state), fake_a_ (support for the next state)."""
        if self.cuda:
            data = to_device(data, self._device)
        s, a, r, s_, d = data['s'], data['a'], data['r'], data['s_'], data['d']
        fake_a, fake_a_ = data['fake_a'], data['fake_a_']

        if self.behavior_policy_stop_training_iter > 0:
            behavior_loss = self._model.score_model_loss_fn(a, s)
            self.behavior_model_optimizer.zero_grad()
            behavior_loss.backward()
            self.behavior_model_optimizer.step()
            self.behavior_policy_stop_training_iter -= 1
            behavior_loss = behavior_loss.item()
        else:
            behavior_loss = 0

        self.energy_guided_policy_begin_training_iter -= 1
        self.q_value_stop_training_iter -= 1
        if self.energy_guided_policy_begin_training_iter < 0:
            if self.q_value_stop_training_iter > 0:
                q0_loss = self._model.q_loss_fn(a, s, r, s_, d, fake_a_, discount=self.discount)
                self.q_optimizer.zero_grad()
                q0_loss.backward()
                self.q_optimizer.step()
                for param, target_param in zip(self._model.q.q0.parameters(),
                                               self._model.q.q0_target.parameters()):
                    target_param.data.copy_(
                        self.qt_update_momentum * param.data + (1 - self.qt_update_momentum) * target_param.data
                    )
                q0_loss = q0_loss.item()
            else:
                q0_loss = 0
            qt_loss = self._model.qt_loss_fn(s, fake_a)
            self.qt_optimizer.zero_grad()
            qt_loss.backward()
            self.qt_optimizer.step()
            qt_loss = qt_loss.item()
        else:
            q0_loss = 0
            qt_loss = 0
        return dict(
            total_loss=behavior_loss + q0_loss + qt_loss,
            behavior_model_training_loss=behavior_loss,
            q0_loss=q0_loss,
            qt_loss=qt_loss,
        )

    def _init_collect(self) -> None:
        pass

    def _forward_collect(self, *args, **kwargs) -> None:
        pass

    def _init_eval(self) -> None:
        self.diffusion_steps = self._cfg.eval.diffusion_steps

    def _forward_eval(self, data: dict, guidance_scale: float = 1.0) -> dict:
        data_id = list(data.keys())
        states = default_collate(list(data.values()))
        actions = self._model.select_actions(
            states, diffusion_steps=self.diffusion_steps, guidance_scale=guidance_scale
        )
        return {i: {'action': d} for i, d in zip(data_id, actions)}

    def _get_train_sample(self, transitions: List[Dict[str, Any]]):
        pass

    def _process_transition(self, *args, **kwargs):
        pass

    def _state_dict_learn(self) -> Dict[str, Any]:
        return {
            'model': self._model.state_dict(),
            'behavior_model_optimizer': self.behavior_model_optimizer.state_dict(),
        }

    def _load_state_dict_learn(self, state_dict: Dict[str, Any]) -> None:
        self._model.load_state_dict(state_dict['model'])
        self.behavior_model_optimizer.load_state_dict(state_dict['behavior_model_optimizer'])

    def _monitor_vars_learn(self) -> List[str]:
        return ['total_loss', 'behavior_model_training_loss', 'q0_loss', 'qt_loss']

    def default_model(self) -> tuple:
        return 'qgpo', ['ding.model.template.qgpo']
