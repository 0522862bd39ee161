"""Multi-discrete action-space policy variants (reference
dizoo/common/policy/md_dqn.py, md_ppo.py, md_rainbow_dqn.py): the model
emits a LIST of per-branch logits/q-values; losses are computed per branch
and averaged. Used by envs whose action is a vector of independent
discrete choices (gym_soccer-style, gobigger).
"""
from typing import Any, Dict, List

import torch

from ding.policy import DQNPolicy, PPOOffPolicy
from ding.policy.common_utils import default_preprocess_learn
from ding.rl_utils import q_nstep_td_data, q_nstep_td_error, ppo_data, ppo_error
from ding.torch_utils import to_device
from ding.utils import POLICY_REGISTRY


@POLICY_REGISTRY.register('md_dqn')
class MultiDiscreteDQNPolicy(DQNPolicy):
    """DQN over a vector of independent discrete action branches."""

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(
            data, use_priority=self._priority, use_priority_IS_weight=self._cfg.priority_IS_weight,
            use_nstep=True, ignore_done=self._cfg.learn.ignore_done,
        )
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        q_value = self._learn_model.forward(data['obs'])['logit']
        with torch.no_grad():
            target_q_value = self._target_model.forward(data['next_obs'])['logit']
            target_q_action = self._learn_model.forward(data['next_obs'])['action']
        if not isinstance(q_value, list):  # single-branch: plain DQN lane
            q_value, target_q_value, target_q_action = [q_value], [target_q_value], [target_q_action]
            actions = [data['action']]
        else:
            act = data['action']
            actions = [act[..., i] if act.dim() > 1 else act for i in range(len(q_value))]
        value_gamma = data.get('value_gamma')
        losses, tds = [], []
        for i, (q, tq, ta) in enumerate(zip(q_value, target_q_value, target_q_action)):
            td_data = q_nstep_td_data(
                q, tq, actions[i].reshape(-1), ta[i] if isinstance(ta, list) else ta,
                data['reward'], data['done'], data['weight']
            )
            loss, td = q_nstep_td_error(td_data, self._gamma, nstep=self._nstep, value_gamma=value_gamma)
            losses.append(loss)
            tds.append(td.abs())
        total = sum(losses) / len(losses)
        priority = torch.stack(tds, dim=-1).mean(-1)
        self._optimizer.zero_grad()
        total.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'cur_lr': self._optimizer.defaults['lr'],
            'total_loss': total.item(),
            'priority': priority.tolist(),
        }


@POLICY_REGISTRY.register('md_ppo')
class MultiDiscretePPOOffPolicy(PPOOffPolicy):
    """Off-policy PPO over independent discrete branches: the clipped
    surrogate is applied per branch and averaged."""

    def _branch_ppo_error(self, output, batch, adv):
        logits_new, logits_old = output['logit'], batch['logit']
        actions = batch['action']
        n = len(logits_new)
        pol_losses, val_losses, ent_losses = [], [], []
        for i in range(n):
            d = ppo_data(
                logits_new[i], logits_old[i] if isinstance(logits_old, list) else logits_old[..., i, :],
                actions[..., i] if actions.dim() > 1 else actions, output['value'], batch['value'], adv,
                batch['return'], batch.get('weight')
            )
            loss, info = ppo_error(d, self._clip_ratio)
            pol_losses.append(loss.policy_loss)
            val_losses.append(loss.value_loss)
            ent_losses.append(loss.entropy_loss)
        return (
            sum(pol_losses) / n, sum(val_losses) / n, sum(ent_losses) / n, info
        )
