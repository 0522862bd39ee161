"""Supervised image-classification policy (reference
dizoo/image_classification/policy/policy.py ImageClassificationPolicy):
plain CE training driven through the BaseLearner + MetricSerialEvaluator
machinery — demonstrates the framework's supervised-learning lane.
"""
from typing import Any, Dict, List

import torch
import torch.nn.functional as F

from ding.model import model_wrap
from ding.policy import Policy
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY


@POLICY_REGISTRY.register('image_classification')
class ImageClassificationPolicy(Policy):

    config = dict(
        type='image_classification',
        cuda=False,
        on_policy=False,
        priority=False,
        model=dict(),
        learn=dict(batch_size=64, learning_rate=0.01, weight_decay=1e-4, update_per_collect=1),
        collect=dict(unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=100, )),
    )

    def default_model(self) -> tuple:
        return 'bc', ['ding.model.template.bc']

    def _init_learn(self) -> None:
        self._optimizer = Adam(
            self._model.parameters(), lr=self._cfg.learn.learning_rate,
            weight_decay=self._cfg.learn.weight_decay
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.reset()

    def _forward_learn(self, data) -> Dict[str, Any]:
        if isinstance(data, (list, tuple)) and isinstance(data[0], (list, tuple)):
            imgs = torch.stack([d[0] for d in data])
            labels = torch.stack([torch.as_tensor(d[1]) for d in data]).reshape(-1)
        else:
            imgs, labels = data['obs'], data['label'].reshape(-1)
        if self._cuda:
            imgs, labels = to_device(imgs, self._device), to_device(labels, self._device)
        self._learn_model.train()
        logit = self._learn_model.forward(imgs)['logit']
        loss = F.cross_entropy(logit, labels.long())
        acc = (logit.argmax(-1) == labels).float().mean()
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        return {'cur_lr': self._optimizer.defaults['lr'], 'total_loss': loss.item(), 'acc': acc.item()}

    def _monitor_vars_learn(self) -> List[str]:
        return ['cur_lr', 'total_loss', 'acc']

    def _init_collect(self) -> None:
        pass

    def _forward_collect(self, data, **kwargs):
        raise NotImplementedError("supervised policy: no env collection")

    def _process_transition(self, obs, policy_output, timestep):
        raise NotImplementedError

    def _get_train_sample(self, data):
        raise NotImplementedError

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='base')
        self._eval_model.reset()

    def _forward_eval(self, data: torch.Tensor) -> Dict[str, Any]:
        """MetricSerialEvaluator contract: batched inputs -> logits."""
        if self._cuda:
            data = to_device(data, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            return self._eval_model.forward(data)
