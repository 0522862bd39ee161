"""Q-learning model family: DQN, BDQ, C51, QRDQN, IQN, FQF, Rainbow, DRQN,
GTrXL-DQN.

Parity: reference ding/model/template/q_learning.py (registrations 'dqn',
'bdq', 'c51dqn', 'qrdqn', 'iqn', 'fqf', 'rainbowdqn', 'drqn', 'gtrxldqn').
"""
from typing import Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ding.torch_utils import get_lstm, GTrXL
from ding.utils import MODEL_REGISTRY, squeeze
from ..common import (
    ConvEncoder, FCEncoder, DiscreteHead, DuelingHead, DistributionHead, RainbowHead, QRDQNHead, QuantileHead,
    FQFHead, BranchingHead, MultiHead,
)


def _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type, dropout=None):
    obs_shape = squeeze(obs_shape)
    if isinstance(obs_shape, int) or len(obs_shape) == 1:
        return FCEncoder(
            squeeze(obs_shape), encoder_hidden_size_list, activation=activation, norm_type=norm_type, dropout=dropout
        )
    if len(obs_shape) == 3:
        return ConvEncoder(obs_shape, encoder_hidden_size_list, activation=activation, norm_type=norm_type)
    raise RuntimeError(f"unsupported obs_shape for the default encoder: {obs_shape}")


@MODEL_REGISTRY.register('dqn')
class DQN(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        dueling: bool = True,
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        dropout: Optional[float] = None,
        init_bias: Optional[float] = None,
        noise: bool = False,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type, dropout)
        head_cls = DuelingHead if dueling else DiscreteHead
        if isinstance(action_shape, int):
            self.head = head_cls(
                head_hidden_size, action_shape, head_layer_num, activation=activation, norm_type=norm_type,
                dropout=dropout, noise=noise
            )
        else:
            self.head = MultiHead(
                head_cls, head_hidden_size, action_shape, layer_num=head_layer_num, activation=activation,
                norm_type=norm_type, dropout=dropout, noise=noise
            )
        if init_bias is not None and not dueling:
            with torch.no_grad():
                last_fc = [m for m in self.head.modules() if isinstance(m, nn.Linear)][-1]
                last_fc.bias.fill_(init_bias)

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))


@MODEL_REGISTRY.register('bdq')
class BDQ(nn.Module):
    """Branching dueling Q network for high-dim discretized action spaces."""

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        num_branches: int = 0,
        action_bins_per_branch: int = 2,
        layer_num: int = 3,
        a_layer_num: Optional[int] = None,
        v_layer_num: Optional[int] = None,
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        norm_type: Optional[str] = None,
        activation=nn.ReLU(),
    ):
        super().__init__()
        obs_shape = squeeze(obs_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type)
        self.num_branches = num_branches
        self.head = BranchingHead(
            head_hidden_size, num_branches=num_branches, action_bins_per_branch=action_bins_per_branch,
            layer_num=layer_num, a_layer_num=a_layer_num, v_layer_num=v_layer_num, activation=activation,
            norm_type=norm_type
        )

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))


@MODEL_REGISTRY.register('c51dqn')
class C51DQN(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        v_min: float = -10,
        v_max: float = 10,
        n_atom: int = 51,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type)
        if isinstance(action_shape, int):
            self.head = DistributionHead(
                head_hidden_size, action_shape, head_layer_num, n_atom=n_atom, v_min=v_min, v_max=v_max,
                activation=activation, norm_type=norm_type
            )
        else:
            self.head = MultiHead(
                DistributionHead, head_hidden_size, action_shape, layer_num=head_layer_num, n_atom=n_atom,
                v_min=v_min, v_max=v_max, activation=activation, norm_type=norm_type
            )

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))


@MODEL_REGISTRY.register('qrdqn')
class QRDQN(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        num_quantiles: int = 32,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type)
        if isinstance(action_shape, int):
            self.head = QRDQNHead(
                head_hidden_size, action_shape, head_layer_num, num_quantiles=num_quantiles, activation=activation,
                norm_type=norm_type
            )
        else:
            self.head = MultiHead(
                QRDQNHead, head_hidden_size, action_shape, layer_num=head_layer_num, num_quantiles=num_quantiles,
                activation=activation, norm_type=norm_type
            )

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))


@MODEL_REGISTRY.register('iqn')
class IQN(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        num_quantiles: int = 32,
        quantile_embedding_size: int = 128,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type)
        if isinstance(action_shape, int):
            self.head = QuantileHead(
                head_hidden_size, action_shape, head_layer_num, num_quantiles=num_quantiles,
                quantile_embedding_size=quantile_embedding_size, activation=activation, norm_type=norm_type
            )
        else:
            self.head = MultiHead(
                QuantileHead, head_hidden_size, action_shape, layer_num=head_layer_num, num_quantiles=num_quantiles,
                quantile_embedding_size=quantile_embedding_size, activation=activation, norm_type=norm_type
            )

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))


@MODEL_REGISTRY.register('fqf')
class FQF(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        num_quantiles: int = 32,
        quantile_embedding_size: int = 128,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type)
        self.head = FQFHead(
            head_hidden_size, action_shape, head_layer_num, num_quantiles=num_quantiles,
            quantile_embedding_size=quantile_embedding_size, activation=activation, norm_type=norm_type
        )

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))


@MODEL_REGISTRY.register('rainbowdqn')
class RainbowDQN(nn.Module):

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        v_min: float = -10,
        v_max: float = 10,
        n_atom: int = 51,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type)
        self.head = RainbowHead(
            head_hidden_size, action_shape, head_layer_num, n_atom=n_atom, v_min=v_min, v_max=v_max,
            activation=activation, norm_type=norm_type, noise=True
        )

    def forward(self, x: torch.Tensor) -> Dict:
        return self.head(self.encoder(x))

    def reset_noise(self):
        from ding.torch_utils import NoisyLinearLayer
        for m in self.modules():
            if isinstance(m, NoisyLinearLayer):
                m.reset_noise()


def parallel_wrapper(forward_fn):
    """Fold [T, B, ...] into [T*B, ...] around a per-frame head call."""

    def wrapper(x):
        T, B = x.shape[:2]
        x = x.reshape(T * B, *x.shape[2:])
        out = forward_fn(x)

        def restore(t):
            if isinstance(t, torch.Tensor):
                return t.reshape(T, B, *t.shape[1:])
            if isinstance(t, list):
                return [restore(i) for i in t]
            if isinstance(t, dict):
                return {k: restore(v) for k, v in t.items()}
            return t

        return restore(out)

    return wrapper


@MODEL_REGISTRY.register('drqn')
class DRQN(nn.Module):
    """DQN + LSTM over time (R2D2 backbone). forward input:
    {'obs': [T,B,...], 'prev_state': ...} -> {'logit', 'next_state'}."""

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        encoder_hidden_size_list: Sequence = [128, 128, 64],
        dueling: bool = True,
        head_hidden_size: Optional[int] = None,
        head_layer_num: int = 1,
        lstm_type: str = 'normal',
        activation=nn.ReLU(),
        norm_type: Optional[str] = None,
        res_link: bool = False,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if head_hidden_size is None:
            head_hidden_size = encoder_hidden_size_list[-1]
        self.encoder = _build_encoder(obs_shape, encoder_hidden_size_list, activation, norm_type)
        self.rnn = get_lstm(lstm_type, head_hidden_size, head_hidden_size)
        self.res_link = res_link
        head_cls = DuelingHead if dueling else DiscreteHead
        if isinstance(action_shape, int):
            self.head = head_cls(head_hidden_size, action_shape, head_layer_num, activation=activation, norm_type=norm_type)
        else:
            self.head = MultiHead(
                head_cls, head_hidden_size, action_shape, layer_num=head_layer_num, activation=activation,
                norm_type=norm_type
            )

    def forward(self, inputs: Dict, inference: bool = False, saved_state_timesteps: Optional[list] = None) -> Dict:
        x, prev_state = inputs['obs'], inputs['prev_state']
        if inference:
            x = self.encoder(x)
            x = x.unsqueeze(0)  # [1, B, H]
            x, next_state = self.rnn(x, prev_state)
            x = x.squeeze(0)
            out = self.head(x)
            out['next_state'] = next_state
            return out
        assert x.dim() in (3, 5), x.shape  # [T, B, C] or [T, B, C, H, W]
        x = parallel_wrapper(self.encoder)(x)  # [T, B, H]
        saved_state = []
        if saved_state_timesteps is not None:
            lstm_embedding = []
            hidden_state_list = []
            cur_state = prev_state
            outputs = []
            for t in range(x.shape[0]):
                output, cur_state = self.rnn(x[t:t + 1], cur_state, list_next_state=False)
                if t + 1 in saved_state_timesteps:
                    # per-sample list for later burn-in reuse
                    h, c = cur_state
                    saved_state.append(
                        [{'h': h[:, i:i + 1], 'c': c[:, i:i + 1]} for i in range(h.shape[1])]
                    )
                outputs.append(output)
            x = torch.cat(outputs, dim=0)
            h, c = cur_state
            next_state = [{'h': h[:, i:i + 1].detach(), 'c': c[:, i:i + 1].detach()} for i in range(h.shape[1])]
        else:
            x, next_state = self.rnn(x, prev_state)
        out = parallel_wrapper(self.head)(x)
        out['next_state'] = next_state
        if saved_state_timesteps is not None:
            out['saved_state'] = saved_state
        return out


@MODEL_REGISTRY.register('gtrxldqn')
class GTrXLDQN(nn.Module):
    """GTrXL encoder + dueling/discrete head with segment memory."""

    def __init__(
        self,
        obs_shape: Union[int, Sequence],
        action_shape: Union[int, Sequence],
        head_layer_num: int = 1,
        att_head_dim: int = 16,
        hidden_size: int = 16,
        att_head_num: int = 2,
        att_mlp_num: int = 2,
        att_layer_num: int = 3,
        memory_len: int = 64,
        activation=nn.ReLU(),
        head_norm_type: Optional[str] = None,
        dropout: float = 0.0,
        gru_gating: bool = True,
        gru_bias: float = 2.0,
        dueling: bool = True,
        encoder_hidden_size_list: Sequence = [128, 128, 256],
        encoder_norm_type: Optional[str] = None,
    ):
        super().__init__()
        obs_shape, action_shape = squeeze(obs_shape), squeeze(action_shape)
        if isinstance(obs_shape, int) or len(obs_shape) == 1:
            input_dim = squeeze(obs_shape)
            self.obs_encoder = None
        else:
            self.obs_encoder = ConvEncoder(
                obs_shape, encoder_hidden_size_list, activation=activation, norm_type=encoder_norm_type
            )
            input_dim = encoder_hidden_size_list[-1]
        self.core = GTrXL(
            input_dim=input_dim, head_dim=att_head_dim, embedding_dim=hidden_size, head_num=att_head_num,
            mlp_num=att_mlp_num, layer_num=att_layer_num, memory_len=memory_len, dropout_ratio=dropout,
            gru_gating=gru_gating, gru_bias=gru_bias
        )
        head_cls = DuelingHead if dueling else DiscreteHead
        self.head = head_cls(hidden_size, action_shape, head_layer_num, activation=activation, norm_type=head_norm_type)

    def forward(self, x: torch.Tensor) -> Dict:
        """x: [T, B, ...]; returns {'logit' [T,B,N], 'memory', 'transformer_out'}."""
        if self.obs_encoder is not None:
            x = parallel_wrapper(self.obs_encoder)(x)
        o = self.core(x)
        out = parallel_wrapper(self.head)(o['logit'])
        out['memory'] = o.get('memory')
        out['transformer_out'] = o['logit']
        return out

    def reset_memory(self, batch_size: Optional[int] = None, state: Optional[torch.Tensor] = None):
        self.core.reset_memory(batch_size, state)

    def get_memory(self):
        return self.core.get_memory()
