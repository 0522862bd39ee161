"""ATOC: attentional communication MARL (DDPG-style QAC + learned
initiator gate + bidirectional-GRU thought integration).

Parity: reference ding/model/template/atoc.py (ATOCAttentionUnit:11,
ATOCCommunicationNet:67, ATOCActorNet:115, ATOC:350). Re-designed: group
selection is a batched top-k over pairwise thought distances (the
reference's per-(b,i) python loop), communication still runs one
bi-GRU pass over the gathered member thoughts.
"""
import copy
from typing import Dict, Optional, Tuple, Union

import torch
import torch.nn as nn

from ding.model.common.head import RegressionHead
from ding.utils import MODEL_REGISTRY, squeeze


def _mlp(in_c: int, hidden: int, out_c: int, layer_num: int = 2, act=nn.ReLU):
    mods = []
    d = in_c
    for _ in range(layer_num - 1):
        mods += [nn.Linear(d, hidden), act()]
        d = hidden
    mods.append(nn.Linear(d, out_c))
    return nn.Sequential(*mods)


class ATOCAttentionUnit(nn.Module):
    """Thought -> initiator probability."""

    def __init__(self, thought_size: int, embedding_size: int):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(thought_size, embedding_size), nn.ReLU(),
            nn.Linear(embedding_size, embedding_size), nn.ReLU(),
            nn.Linear(embedding_size, 1),
        )

    def forward(self, thoughts: torch.Tensor) -> torch.Tensor:
        return torch.sigmoid(self.net(thoughts)).squeeze(-1)


class ATOCCommunicationNet(nn.Module):
    """Bidirectional GRU integrating thoughts within a group."""

    def __init__(self, thought_size: int):
        super().__init__()
        assert thought_size % 2 == 0
        self.gru = nn.GRU(thought_size, thought_size // 2, bidirectional=True)

    def forward(self, thoughts: torch.Tensor) -> torch.Tensor:
        """thoughts: [group_size, n_groups, T] -> integrated same shape."""
        out, _ = self.gru(thoughts)
        return out


class ATOCActorNet(nn.Module):

    def __init__(
        self,
        obs_shape: Union[Tuple, int],
        thought_size: int,
        action_shape: int,
        n_agent: int,
        communication: bool = True,
        agent_per_group: int = 2,
        initiator_threshold: float = 0.5,
        attention_embedding_size: int = 64,
        actor_1_embedding_size: Optional[int] = None,
        actor_2_embedding_size: Optional[int] = None,
    ):
        super().__init__()
        self._obs_shape = squeeze(obs_shape)
        self._thought_size = thought_size
        self._act_shape = action_shape
        self._n_agent = n_agent
        self._communication = communication
        self._agent_per_group = agent_per_group
        self._initiator_threshold = initiator_threshold
        a1 = actor_1_embedding_size or thought_size
        a2 = actor_2_embedding_size or thought_size
        self.actor_1 = _mlp(self._obs_shape, a1, thought_size)
        self.actor_2 = nn.Sequential(
            nn.Linear(thought_size * 2, a2), nn.ReLU(),
            RegressionHead(a2, self._act_shape, 2, final_tanh=True),
        )
        if communication:
            self.attention = ATOCAttentionUnit(thought_size, attention_embedding_size)
            self.comm_net = ATOCCommunicationNet(thought_size)

    def forward(self, obs: torch.Tensor) -> Dict:
        assert obs.dim() == 3 and obs.shape[1] == self._n_agent
        current_thoughts = self.actor_1(obs)  # [B, A, T]
        if self._communication:
            old_thoughts = current_thoughts.clone().detach()
            init_prob, is_initiator, group = self._get_initiate_group(old_thoughts)
            new_thoughts = self._get_new_thoughts(current_thoughts, group, is_initiator)
        else:
            new_thoughts = current_thoughts
        action = self.actor_2(torch.cat([current_thoughts, new_thoughts], dim=-1))['pred']
        if self._communication:
            return {
                'action': action, 'group': group, 'initiator_prob': init_prob,
                'is_initiator': is_initiator, 'new_thoughts': new_thoughts, 'old_thoughts': old_thoughts,
            }
        return {'action': action}

    def _get_initiate_group(self, thoughts: torch.Tensor):
        """Initiators pick their agent_per_group nearest thoughts (batched
        top-k over the pairwise squared-distance matrix)."""
        init_prob = self.attention(thoughts)  # [B, A]
        is_initiator = init_prob > self._initiator_threshold
        B, A = init_prob.shape
        dot = thoughts.bmm(thoughts.transpose(1, 2))
        sq = dot.diagonal(0, 1, 2)
        dists = sq.unsqueeze(1) - 2 * dot + sq.unsqueeze(2)  # [B, A, A]
        k = min(self._agent_per_group, A)
        nearest = dists.topk(k, dim=-1, largest=False).indices  # [B, A, k]
        group = torch.zeros(B, A, A, device=thoughts.device)
        group.scatter_(2, nearest, 1.0)
        group = group * is_initiator.unsqueeze(-1).float()  # only initiators form groups
        return init_prob, is_initiator, group

    def _get_new_thoughts(self, current_thoughts: torch.Tensor, group: torch.Tensor, is_initiator: torch.Tensor):
        B, A = current_thoughts.shape[:2]
        new_thoughts = current_thoughts.detach().clone()
        idx = is_initiator.nonzero(as_tuple=False)
        if idx.numel() == 0:
            return new_thoughts
        # gather member thoughts per initiator -> [k, n_groups, T]
        members = []
        slots = []
        for b, i in idx.tolist():
            mem = group[b, i].nonzero(as_tuple=False).squeeze(-1)
            members.append(new_thoughts[b, mem])
            slots.append((b, mem))
        k = max(m.shape[0] for m in members)
        stacked = torch.stack([
            torch.cat([m, m.new_zeros(k - m.shape[0], m.shape[1])]) if m.shape[0] < k else m for m in members
        ], dim=1)  # [k, n_groups, T]
        integrated = self.comm_net(stacked)
        for g, (b, mem) in enumerate(slots):
            new_thoughts[b, mem] = integrated[:len(mem), g]
        return new_thoughts


@MODEL_REGISTRY.register('atoc')
class ATOC(nn.Module):
    """ATOC QAC: communicating actor + per-agent critic Q(o_i, a_i)."""

    mode = ['compute_actor', 'compute_critic', 'optimize_actor_attention']

    def __init__(
        self,
        obs_shape: Union[int, Tuple],
        action_shape: Union[int, Tuple],
        thought_size: int,
        n_agent: int,
        communication: bool = True,
        agent_per_group: int = 2,
        actor_1_embedding_size: Optional[int] = None,
        actor_2_embedding_size: Optional[int] = None,
        critic_head_hidden_size: int = 64,
        critic_head_layer_num: int = 2,
        **kwargs,
    ):
        super().__init__()
        self._communication = communication
        obs_shape = squeeze(obs_shape)
        action_shape = squeeze(action_shape)
        self.actor = ATOCActorNet(
            obs_shape, thought_size, action_shape, n_agent, communication, agent_per_group,
            actor_1_embedding_size=actor_1_embedding_size, actor_2_embedding_size=actor_2_embedding_size,
        )
        self.critic = nn.Sequential(
            nn.Linear(obs_shape + action_shape, critic_head_hidden_size), nn.ReLU(),
            RegressionHead(critic_head_hidden_size, 1, critic_head_layer_num, final_tanh=False),
        )

    def _compute_delta_q(self, obs: torch.Tensor, actor_outputs: Dict) -> torch.Tensor:
        """Initiator credit: mean Q-gain of its group from communication."""
        assert obs.dim() == 3
        new_thoughts = actor_outputs['new_thoughts']
        old_thoughts = actor_outputs['old_thoughts']
        group = actor_outputs['group']
        is_initiator = actor_outputs['is_initiator']
        B, A = new_thoughts.shape[:2]
        delta_q = torch.zeros(B, A, device=new_thoughts.device)
        with torch.no_grad():
            idx = is_initiator.nonzero(as_tuple=False)
            for b, i in idx.tolist():
                mem = group[b, i].nonzero(as_tuple=False).squeeze(-1)
                if mem.numel() == 0:
                    continue
                before_a = self.actor.actor_2(
                    torch.cat([old_thoughts[b, mem], old_thoughts[b, mem]], dim=-1)
                )['pred']
                after_a = self.actor.actor_2(
                    torch.cat([old_thoughts[b, mem], new_thoughts[b, mem]], dim=-1)
                )['pred']
                q_before = self.critic(torch.cat([obs[b, mem], before_a], dim=-1))['pred']
                q_after = self.critic(torch.cat([obs[b, mem], after_a], dim=-1))['pred']
                delta_q[b, i] = q_after.mean() - q_before.mean()
        return delta_q

    def compute_actor(self, obs: torch.Tensor, get_delta_q: bool = False) -> Dict[str, torch.Tensor]:
        outputs = self.actor(obs)
        if get_delta_q and self._communication:
            outputs['delta_q'] = self._compute_delta_q(obs, outputs)
        return outputs

    def compute_critic(self, inputs: Dict) -> Dict:
        obs, action = inputs['obs'], inputs['action']
        if action.dim() == 2:
            action = action.unsqueeze(2)
        q = self.critic(torch.cat([obs, action], dim=-1))['pred']
        return {'q_value': q}

    def optimize_actor_attention(self, inputs: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        """BCE of initiator prob against normalized delta_q."""
        if not self._communication:
            raise NotImplementedError
        delta_q = inputs['delta_q'].reshape(-1)
        init_prob = inputs['initiator_prob'].reshape(-1)
        is_init = inputs['is_initiator'].reshape(-1)
        sel = is_init.nonzero(as_tuple=False).squeeze(-1)
        if sel.numel() == 0:
            loss = torch.zeros((), device=delta_q.device, requires_grad=True)
            return {'loss': loss}
        delta_q = delta_q[sel]
        init_prob = 0.9 * init_prob[sel] + 0.05
        loss = -delta_q * torch.log(init_prob) - (1 - delta_q) * torch.log(1 - init_prob)
        return {'loss': loss.mean()}

    def forward(self, inputs, mode: str, **kwargs) -> Dict:
        assert mode in self.mode, mode
        return getattr(self, mode)(inputs, **kwargs)
