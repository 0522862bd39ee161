"""Remaining registered model templates: SQN (twin discrete Q), PPG
(actor-critic + aux value head), BCQ (VAE behavior + perturbation actor +
twin critic), NGU recurrent Q, procedure-cloning nets, autoregressive EBM,
ContinuousQVAC.

Parity: reference ding/model/template/sqn.py ('sqn':10), ppg.py ('ppg':10),
bcq.py ('bcq':13), ngu.py ('ngu':44), procedure_cloning.py
('pc_mcts':80, 'pc_bfs':267), ebm.py AutoregressiveEBM ('arebm':803),
qvac.py ('continuous_qvac':13).
"""
import copy
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.torch_utils import get_lstm
from ding.utils import MODEL_REGISTRY, squeeze
from ..common.encoder import ConvEncoder, FCEncoder
from ..common.head import DuelingHead, DiscreteHead, RegressionHead, ReparameterizationHead
from .q_learning import DQN
from .vac import VAC
from .vae import VanillaVAE
from .ebm import EBM


@MODEL_REGISTRY.register('sqn')
class SQN(nn.Module):
    """Twin discrete Q networks (soft Q-learning for discrete actions)."""

    def __init__(self, *args, **kwargs):
        super().__init__()
        self.q0 = DQN(*args, **kwargs)
        self.q1 = DQN(*args, **kwargs)

    def forward(self, data: torch.Tensor) -> Dict:
        out0 = self.q0(data)
        out1 = self.q1(data)
        return {'q_value': [out0['logit'], out1['logit']], 'logit': out0['logit']}


@MODEL_REGISTRY.register('ppg')
class PPG(nn.Module):
    """VAC + detached auxiliary value head (phasic policy gradient)."""

    mode = ['compute_actor', 'compute_critic', 'compute_actor_critic']

    def __init__(self, obs_shape, action_shape, action_space: str = 'discrete', share_encoder: bool = True,
                 encoder_hidden_size_list=[128, 128, 64], actor_head_hidden_size: int = 64,
                 actor_head_layer_num: int = 2, critic_head_hidden_size: int = 64, critic_head_layer_num: int = 1,
                 **kwargs):
        super().__init__()
        self.actor_critic = VAC(
            obs_shape, action_shape, action_space=action_space, share_encoder=share_encoder,
            encoder_hidden_size_list=encoder_hidden_size_list, actor_head_hidden_size=actor_head_hidden_size,
            actor_head_layer_num=actor_head_layer_num, critic_head_hidden_size=critic_head_hidden_size,
            critic_head_layer_num=critic_head_layer_num,
        )
        self.aux_critic = copy.deepcopy(self.actor_critic.critic)

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode, mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, x) -> Dict:
        return self.actor_critic(x, mode='compute_actor')

    def compute_critic(self, x) -> Dict:
        return self.actor_critic(x, mode='compute_critic')

    def compute_actor_critic(self, x) -> Dict:
        return self.actor_critic(x, mode='compute_actor_critic')


@MODEL_REGISTRY.register('bcq')
class BCQ(nn.Module):
    """BCQ: VAE models the behavior distribution; the actor perturbs VAE
    actions within phi; twin critics rank candidates at eval."""

    mode = ['compute_actor', 'compute_critic', 'compute_vae', 'compute_eval']

    def __init__(
        self,
        obs_shape: int,
        action_shape: int,
        actor_head_hidden_size: List = [400, 300],
        critic_head_hidden_size: List = [400, 300],
        vae_hidden_dims: List = [750, 750],
        phi: float = 0.05,
        **kwargs,
    ):
        super().__init__()
        obs_shape = squeeze(obs_shape)
        action_shape = squeeze(action_shape)
        self.action_shape = action_shape
        self.phi = phi

        def mlp(dims_in, hidden):
            net, d = [], dims_in
            for h in hidden:
                net += [nn.Linear(d, h), nn.ReLU()]
                d = h
            net.append(nn.Linear(d, 1))
            return nn.Sequential(*net)

        cin = obs_shape + action_shape
        self.critic = nn.ModuleList([mlp(cin, critic_head_hidden_size) for _ in range(2)])
        self.actor = mlp(cin, actor_head_hidden_size)
        self.vae = VanillaVAE(action_shape, obs_shape, action_shape * 2, vae_hidden_dims)

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode, mode
        return getattr(self, mode)(inputs)

    def compute_critic(self, inputs: Dict) -> Dict:
        obs, action = inputs['obs'], inputs['action']
        if action.dim() == 1:
            action = action.unsqueeze(1)
        x = torch.cat([obs, action], dim=-1)
        return {'q_value': [m(x).squeeze(-1) for m in self.critic]}

    def compute_actor(self, inputs: Dict) -> Dict:
        x = self.actor(torch.cat([inputs['obs'], inputs['action']], dim=-1))
        action = self.phi * torch.tanh(x)
        action = (action + inputs['action']).clamp(-1, 1)
        return {'action': action}

    def compute_vae(self, inputs: Dict) -> Dict:
        return self.vae.forward(inputs)

    def compute_eval(self, inputs: Dict) -> Dict:
        obs = inputs['obs']
        obs_rep = obs.unsqueeze(0).repeat_interleave(100, dim=0)
        z = torch.randn(obs_rep.shape[0], obs_rep.shape[1], self.action_shape * 2, device=obs.device)
        z = z.clamp(-0.5, 0.5)
        sample_action = self.vae.decode_with_obs({'z': z, 'obs': obs_rep})['reconstruction_action']
        action = self.compute_actor({'obs': obs_rep, 'action': sample_action})['action']
        q = self.compute_critic({'obs': obs_rep, 'action': action})['q_value'][0]
        idx = q.argmax(dim=0).reshape(1, -1, 1).repeat_interleave(action.shape[-1], dim=-1)
        return {'action': action.gather(0, idx).squeeze(0)}


@MODEL_REGISTRY.register('ngu')
class NGU(nn.Module):
    """R2D2-style recurrent dueling Q whose LSTM input is
    [obs_embedding, prev_action_onehot, prev_extrinsic_reward, beta_onehot]."""

    def __init__(
        self,
        obs_shape,
        action_shape,
        encoder_hidden_size_list=[128, 128, 64],
        collector_env_num: int = 1,
        dueling: bool = True,
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        lstm_type: str = 'normal',
        activation: str = 'relu',
        norm_type: Optional[str] = None,
        **kwargs,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        self.action_shape = action_shape
        self.collector_env_num = collector_env_num
        head_hidden_size = head_hidden_size or encoder_hidden_size_list[-1]
        if isinstance(obs_shape, int) or len(obs_shape) == 1:
            self.encoder = FCEncoder(obs_shape, encoder_hidden_size_list)
        else:
            self.encoder = ConvEncoder(obs_shape, encoder_hidden_size_list)
        input_size = head_hidden_size + action_shape + 1 + collector_env_num
        self.rnn = get_lstm(lstm_type, input_size=input_size, hidden_size=head_hidden_size)
        head_cls = DuelingHead if dueling else DiscreteHead
        self.head = head_cls(head_hidden_size, action_shape, head_layer_num)

    def forward(self, inputs: Dict, inference: bool = False, saved_state_timesteps=None) -> Dict:
        """inputs: {obs [T,B,...] or [B,...], prev_state, prev_action [T,B],
        prev_reward_extrinsic [T,B], beta [T,B]}."""
        x, prev_state = inputs['obs'], inputs['prev_state']
        if inference:
            x = self.encoder(x)
            prev_action = inputs['prev_action']
            prev_reward = inputs['prev_reward_extrinsic']
            beta = inputs['beta']
            a_onehot = F.one_hot(prev_action.long(), self.action_shape).float()
            b_onehot = F.one_hot(beta.long(), self.collector_env_num).float()
            x = torch.cat([x, a_onehot, prev_reward.unsqueeze(-1).float(), b_onehot], dim=-1)
            x = x.unsqueeze(0)
            x, next_state = self.rnn(x, prev_state)
            x = x.squeeze(0)
            out = self.head(x)
            out['next_state'] = next_state
            return out
        T, B = x.shape[:2]
        x = self.encoder(x.reshape(T * B, *x.shape[2:])).reshape(T, B, -1)
        a_onehot = F.one_hot(inputs['prev_action'].long(), self.action_shape).float()
        b_onehot = F.one_hot(inputs['beta'].long(), self.collector_env_num).float()
        r = inputs['prev_reward_extrinsic']
        if r.dim() == 2:
            r = r.unsqueeze(-1)
        x = torch.cat([x, a_onehot, r.float(), b_onehot], dim=-1)
        lstm_out, next_state = self.rnn(x, prev_state)
        outs = self.head(lstm_out.reshape(T * B, -1))
        return {
            'logit': outs['logit'].reshape(T, B, -1),
            'next_state': next_state,
        }


class BFSConvEncoder(nn.Module):
    """Stride-1 'same' conv stack keeping H, W (BFS map -> per-cell logits)."""

    def __init__(self, obs_shape, hidden_size_list, kernel_size, stride, padding):
        super().__init__()
        layers = []
        in_c = obs_shape[0]
        for i, out_c in enumerate(hidden_size_list):
            layers.append(nn.Conv2d(in_c, out_c, kernel_size[i], stride[i], padding[i]))
            if i != len(hidden_size_list) - 1:
                layers.append(nn.ReLU())
            in_c = out_c
        self.main = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.main(x)


@MODEL_REGISTRY.register('pc_bfs')
class ProcedureCloningBFS(nn.Module):
    """Per-cell action-logit map for BFS-style procedure cloning."""

    def __init__(self, obs_shape, action_shape: int, encoder_hidden_size_list=[128, 128, 256, 256]):
        super().__init__()
        hidden = list(encoder_hidden_size_list) + [action_shape + 1]
        n = len(hidden)
        self._encoder = BFSConvEncoder(
            obs_shape=obs_shape, hidden_size_list=hidden, kernel_size=(3, ) * n, stride=(1, ) * n,
            padding=(1, ) * n,
        )

    def forward(self, x: torch.Tensor) -> Dict:
        x = x.permute(0, 3, 1, 2).float() if x.shape[-1] <= 8 and x.dim() == 4 and x.shape[1] > 8 else x.float()
        logits = self._encoder(x)
        return {'logit': logits.permute(0, 2, 3, 1)}


@MODEL_REGISTRY.register('pc_mcts')
class ProcedureCloningMCTS(nn.Module):
    """Observation encoder + causal transformer over intermediate MCTS
    computation tokens, predicting the action sequence."""

    def __init__(
        self,
        obs_shape,
        action_dim: int,
        cnn_hidden_list=[128, 128, 256, 256, 256],
        mlp_hidden_list=[256, 256],
        att_heads: int = 8,
        att_hidden: int = 128,
        n_att: int = 4,
        drop_p: float = 0.1,
        max_T: int = 17,
        **kwargs,
    ):
        super().__init__()
        self.action_dim = action_dim
        self.max_T = max_T
        if isinstance(obs_shape, int) or len(obs_shape) == 1:
            self.obs_encoder = FCEncoder(squeeze(obs_shape), mlp_hidden_list + [att_hidden])
        else:
            self.obs_encoder = ConvEncoder(
                obs_shape, list(cnn_hidden_list[:3]) + [att_hidden],
                kernel_size=[3, 3, 3], stride=[2, 2, 2], padding=[1, 1, 1],
            )
        self.action_embed = nn.Embedding(action_dim, att_hidden)
        self.pos_embed = nn.Parameter(torch.zeros(1, max_T, att_hidden))
        layer = nn.TransformerEncoderLayer(
            att_hidden, att_heads, dim_feedforward=4 * att_hidden, dropout=drop_p, batch_first=True
        )
        self.transformer = nn.TransformerEncoder(layer, n_att)
        self.predict = nn.Linear(att_hidden, action_dim)

    def forward(self, obs: torch.Tensor, actions: torch.Tensor) -> Dict:
        """obs [B, ...], actions [B, T] (teacher-forced seq) -> logits [B, T, A]."""
        B, T = actions.shape
        obs_tok = self.obs_encoder(obs.float()).unsqueeze(1)  # [B, 1, H]
        act_tok = self.action_embed(actions.long())  # [B, T, H]
        toks = torch.cat([obs_tok, act_tok], dim=1)[:, :self.max_T]
        toks = toks + self.pos_embed[:, :toks.shape[1]]
        mask = torch.triu(torch.ones(toks.shape[1], toks.shape[1], device=toks.device, dtype=torch.bool), 1)
        h = self.transformer(toks, mask=mask)
        return {'logit': self.predict(h)[:, :T]}


@MODEL_REGISTRY.register('arebm')
class AutoregressiveEBM(nn.Module):
    """Per-action-dimension EBM chain for autoregressive implicit BC."""

    def __init__(self, obs_shape: int, action_shape: int, hidden_size: int = 512, hidden_layer_num: int = 4):
        super().__init__()
        self.ebm_list = nn.ModuleList(
            [EBM(obs_shape, i + 1, hidden_size, hidden_layer_num) for i in range(action_shape)]
        )

    def forward(self, obs: torch.Tensor, action: torch.Tensor) -> torch.Tensor:
        return torch.stack([ebm(obs, action[..., :i + 1]) for i, ebm in enumerate(self.ebm_list)], dim=-1)


@MODEL_REGISTRY.register('continuous_qvac')
class ContinuousQVAC(nn.Module):
    """QAC with an extra state-value branch (Q, V, actor) for algorithms
    needing both (e.g. IQL-style actor-critic)."""

    mode = ['compute_actor', 'compute_critic']

    def __init__(
        self,
        obs_shape: int,
        action_shape: int,
        action_space: str = 'reparameterization',
        twin_critic: bool = True,
        actor_head_hidden_size: int = 64,
        actor_head_layer_num: int = 1,
        critic_head_hidden_size: int = 64,
        critic_head_layer_num: int = 1,
        **kwargs,
    ):
        super().__init__()
        obs_shape = squeeze(obs_shape)
        action_shape = squeeze(action_shape)
        self.action_space = action_space
        if action_space == 'regression':
            self.actor = nn.Sequential(
                nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
                RegressionHead(actor_head_hidden_size, action_shape, actor_head_layer_num, final_tanh=True),
            )
        else:
            self.actor = nn.Sequential(
                nn.Linear(obs_shape, actor_head_hidden_size), nn.ReLU(),
                ReparameterizationHead(
                    actor_head_hidden_size, action_shape, actor_head_layer_num, sigma_type='conditioned'
                ),
            )
        self.twin_critic = twin_critic
        cin = obs_shape + action_shape

        def q_net():
            return nn.Sequential(
                nn.Linear(cin, critic_head_hidden_size), nn.ReLU(),
                RegressionHead(critic_head_hidden_size, 1, critic_head_layer_num, final_tanh=False),
            )

        self.critic = nn.ModuleList([q_net() for _ in range(2)]) if twin_critic else q_net()
        self.value = nn.Sequential(
            nn.Linear(obs_shape, critic_head_hidden_size), nn.ReLU(),
            RegressionHead(critic_head_hidden_size, 1, critic_head_layer_num, final_tanh=False),
        )

    def forward(self, inputs, mode: str) -> Dict:
        assert mode in self.mode, mode
        return getattr(self, mode)(inputs)

    def compute_actor(self, obs: torch.Tensor) -> Dict:
        x = self.actor(obs)
        if self.action_space == 'regression':
            return {'action': x['pred']}
        return {'logit': [x['mu'], x['sigma']]}

    def compute_critic(self, inputs: Dict) -> Dict:
        obs, action = inputs['obs'], inputs['action']
        if action.dim() == 1:
            action = action.unsqueeze(1)
        x = torch.cat([obs, action], dim=-1)
        if self.twin_critic:
            q = [m(x)['pred'] for m in self.critic]
        else:
            q = self.critic(x)['pred']
        return {'q_value': q, 'v_value': self.value(obs)['pred']}
