"""COMA actor-critic model. Parity: reference ding/model/template/coma.py."""
from typing import Dict, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze
from ding.torch_utils import MLP, one_hot


@MODEL_REGISTRY.register('coma')
class COMA(nn.Module):
    """Actor: per-agent MLP over agent_state. Critic: centralized
    Q(s, a_{-i}, i) over global_state + agent one-hot + other agents'
    actions, output per-action Q [T,B,A,N]."""

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        agent_num: int,
        obs_shape: dict,
        action_shape: Union[int, tuple],
        actor_hidden_size_list=(64, 64),
    ):
        super().__init__()
        action_shape = squeeze(action_shape)
        self.agent_num = agent_num
        self.action_shape = action_shape
        agent_obs = squeeze(obs_shape['agent_state'])
        global_obs = squeeze(obs_shape['global_state'])
        self.actor = MLP(agent_obs, actor_hidden_size_list[0], action_shape, len(actor_hidden_size_list) + 1,
                         activation='relu', output_activation=False, output_norm=False)
        critic_in = global_obs + agent_obs + agent_num + agent_num * action_shape
        self.critic = MLP(critic_in, 128, action_shape, 3, activation='relu', output_activation=False,
                          output_norm=False)

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode
        return getattr(self, f'_{mode}')(inputs)

    def _compute_actor(self, inputs) -> Dict:
        if isinstance(inputs, dict) and 'obs' in inputs:
            obs = inputs['obs']
        else:
            obs = inputs
        agent_state = obs['agent_state']  # [..., A, obs]
        logit = self.actor(agent_state)
        mask = obs.get('action_mask', None)
        if mask is not None:
            logit = logit.masked_fill(~mask.bool(), -9999999)
        return {'logit': logit}

    def _compute_critic(self, inputs) -> Dict:
        obs, action = inputs['obs'], inputs['action']  # action [T,B,A] or [B,A]
        agent_state = obs['agent_state']
        global_state = obs['global_state']
        shape = action.shape  # [..., A]
        A, N = self.agent_num, self.action_shape
        act_oh = one_hot(action.reshape(-1), N).reshape(*shape, N)  # [..., A, N]
        # other agents' actions: zero own slot
        act_all = act_oh.unsqueeze(-3).expand(*shape[:-1], A, A, N).reshape(*shape[:-1], A, A * N).clone()
        # zero out own action block per agent i
        for i in range(A):
            act_all[..., i, i * N:(i + 1) * N] = 0
        agent_id = one_hot(torch.arange(A, device=action.device), A)  # [A, A]
        agent_id = agent_id.expand(*shape[:-1], A, A)
        gs = global_state.unsqueeze(-2).expand(*global_state.shape[:-1], A, global_state.shape[-1])
        x = torch.cat([gs, agent_state, agent_id, act_all], dim=-1)
        q = self.critic(x)  # [..., A, N]
        return {'q_value': q}
