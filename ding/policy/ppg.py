"""PPG: phasic policy gradient (policy phase + auxiliary value phase).

Parity: reference ding/policy/ppg.py ('ppg', 'ppg_offpolicy').
"""
import copy
from collections import deque, namedtuple
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.rl_utils import ppo_data, ppo_error, ppg_data, ppg_joint_error
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .ppo import PPOPolicy


@POLICY_REGISTRY.register('ppg_offpolicy')
class PPGOffPolicy(PPOPolicy):
    """PPO policy phase + periodic joint auxiliary phase distilling value
    into the policy network."""

    config = dict(
        type='ppg_offpolicy',
        on_policy=False,
        recompute_adv=False,
        learn=dict(
            update_per_collect=5,
            batch_size=64,
            learning_rate=3e-4,
            epoch_per_collect=1,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=False,
            aux_freq=5,  # auxiliary phase every N policy updates
            aux_train_epoch=3,
            beta_weight=1.0,
        ),
        other=dict(replay_buffer=dict(replay_buffer_size=10000, ), ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        self._aux_memory = deque(maxlen=2048)
        self._train_count = 0
        self._aux_freq = self._cfg.learn.aux_freq
        self._beta_weight = self._cfg.learn.beta_weight

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        from .common_utils import default_preprocess_learn
        collated = default_preprocess_learn(data, ignore_done=self._cfg.learn.ignore_done, use_nstep=False)
        if self._cuda:
            collated = to_device(collated, self._device)
        self._learn_model.train()
        collated['return'] = collated['adv'] + collated['value']
        output = self._learn_model.forward(collated['obs'], mode='compute_actor_critic')
        adv = collated['adv']
        if self._adv_norm:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        loss, info = ppo_error(
            ppo_data(
                output['logit'], collated['logit'], collated['action'], output['value'], collated['value'], adv,
                collated['return'], collated.get('weight')
            ), self._clip_ratio
        )
        total = loss.policy_loss + self._value_weight * loss.value_loss - self._entropy_weight * loss.entropy_loss
        self._optimizer.zero_grad()
        total.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        # store for the aux phase
        for i in range(collated['obs'].shape[0] if isinstance(collated['obs'], torch.Tensor) else 0):
            self._aux_memory.append({
                'obs': collated['obs'][i].detach(),
                'return': collated['return'][i].detach(),
                'logit': output['logit'][i].detach(),
            })
        self._train_count += 1
        ret = {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': total.item(),
            'policy_loss': loss.policy_loss.item(),
            'value_loss': loss.value_loss.item(),
            'entropy_loss': loss.entropy_loss.item(),
            'aux_value_loss': 0.0,
            'auxiliary_loss': 0.0,
            'behavioral_cloning_loss': 0.0,
        }
        if self._train_count % self._aux_freq == 0 and len(self._aux_memory) >= self._cfg.learn.batch_size:
            aux = self._aux_phase()
            ret.update(aux)
        return ret

    def _aux_phase(self) -> Dict[str, float]:
        import random
        bs = self._cfg.learn.batch_size
        stats = {'auxiliary_loss': 0.0, 'behavioral_cloning_loss': 0.0}
        for _ in range(self._cfg.learn.aux_train_epoch):
            batch = random.sample(list(self._aux_memory), bs)
            obs = torch.stack([b['obs'] for b in batch])
            ret = torch.stack([b['return'] for b in batch])
            old_logit = torch.stack([b['logit'] for b in batch])
            out = self._learn_model.forward(obs, mode='compute_actor_critic')
            joint = ppg_joint_error(
                ppg_data(out['logit'], old_logit, None, out['value'], out['value'].detach(), ret, None),
                self._clip_ratio
            )
            loss = joint.auxiliary_loss + self._beta_weight * joint.behavioral_cloning_loss
            self._optimizer.zero_grad()
            loss.backward()
            self._optimizer.step()
            stats['auxiliary_loss'] += joint.auxiliary_loss.item()
            stats['behavioral_cloning_loss'] += joint.behavioral_cloning_loss.item()
        return stats

    def _monitor_vars_learn(self) -> List[str]:
        return super()._monitor_vars_learn() + ['auxiliary_loss', 'behavioral_cloning_loss']


@POLICY_REGISTRY.register('ppg')
class PPGPolicy(PPGOffPolicy):
    config = dict(type='ppg', on_policy=True)
