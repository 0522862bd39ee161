"""Recurrent cells: LayerNorm-LSTM (custom), plain torch LSTM/GRU wrappers,
and hidden-state (de)serialization helpers.

Parity: reference ding/torch_utils/network/rnn.py (LSTM:131, get_lstm,
sequence_mask). The custom LSTM keeps the two gate GEMMs on rocBLAS
(torch.matmul) — MI355X library GEMMs — while the pointwise
LN+gate+state update is a single fused region per step; on GPU the fused
pointwise op is a HIP kernel candidate (ding/ops, DI-hpc kernel #3 in
SURVEY §2.9a).
"""
from typing import List, Optional, Tuple, Union

import torch
import torch.nn as nn

from ding.torch_utils.data_helper import zeros_like


def is_sequence(data):
    return isinstance(data, (list, tuple))


def sequence_mask(lengths: torch.Tensor, max_len: Optional[int] = None) -> torch.BoolTensor:
    """lengths [B] -> bool mask [B, max_len]."""
    if len(lengths.shape) == 1:
        lengths = lengths.unsqueeze(-1)
    bz = lengths.numel()
    if max_len is None:
        max_len = int(lengths.max())
    mask = torch.arange(max_len, device=lengths.device).view(1, -1) < lengths.view(bz, 1)
    return mask


class LSTMForwardWrapper:
    """Normalize prev_state handling shared by all LSTM variants: accept
    None / list-of-per-sample states / (h, c) tuples; emit per-sample list."""

    def _before_forward(self, inputs: torch.Tensor, prev_state: Union[None, List[dict]]):
        seq_len, batch_size = inputs.shape[:2]
        if prev_state is None:
            zeros = torch.zeros(
                self.num_layers, batch_size, self.hidden_size, dtype=inputs.dtype, device=inputs.device
            )
            return (zeros, zeros.clone())
        if isinstance(prev_state, (list, tuple)) and len(prev_state) == batch_size and \
                (prev_state[0] is None or isinstance(prev_state[0], (dict, type(None)))):
            # per-env states from collector: list of None / {'h':..., 'c':...}
            state = []
            for p in prev_state:
                if p is None:
                    state.append({
                        'h': torch.zeros(self.num_layers, 1, self.hidden_size, dtype=inputs.dtype, device=inputs.device),
                        'c': torch.zeros(self.num_layers, 1, self.hidden_size, dtype=inputs.dtype, device=inputs.device),
                    })
                else:
                    state.append(p)
            h = torch.cat([s['h'] for s in state], dim=1)
            c = torch.cat([s['c'] for s in state], dim=1)
            return (h, c)
        if isinstance(prev_state, (list, tuple)) and len(prev_state) == 2 and isinstance(prev_state[0], torch.Tensor):
            return tuple(prev_state)
        raise TypeError(f"unsupported prev_state: {type(prev_state)}")

    def _after_forward(self, next_state: Tuple[torch.Tensor, torch.Tensor], list_next_state: bool = False):
        if list_next_state:
            h, c = next_state
            batch_size = h.shape[1]
            return [
                {'h': h[:, i:i + 1].detach(), 'c': c[:, i:i + 1].detach()} for i in range(batch_size)
            ]
        return next_state


class LSTM(nn.Module, LSTMForwardWrapper):
    """Multi-layer LSTM with LayerNorm on the pre-activation gate sums.

    Gate math per step (layer l):
      g = LN_x(x W_x^T) + LN_h(h U_h^T) + b      (g: [B, 4H])
      i, f, o, u = split(g); c' = sig(f) c + sig(i) tanh(u); h' = sig(o) tanh(c')
    """

    def __init__(
        self,
        input_size: int,
        hidden_size: int,
        num_layers: int,
        norm_type: Optional[str] = 'LN',
        dropout: float = 0.0,
    ):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.num_layers = num_layers
        dims = [input_size] + [hidden_size] * num_layers
        self.wx = nn.ParameterList()
        self.wh = nn.ParameterList()
        self.bias = nn.ParameterList()
        for l in range(num_layers):
            self.wx.append(nn.Parameter(torch.empty(dims[l], 4 * hidden_size)))
            self.wh.append(nn.Parameter(torch.empty(hidden_size, 4 * hidden_size)))
            self.bias.append(nn.Parameter(torch.zeros(4 * hidden_size)))
        if norm_type == 'LN':
            self.norm_x = nn.ModuleList([nn.LayerNorm(4 * hidden_size) for _ in range(num_layers)])
            self.norm_h = nn.ModuleList([nn.LayerNorm(4 * hidden_size) for _ in range(num_layers)])
        else:
            self.norm_x = self.norm_h = None
        self.use_dropout = dropout > 0.0
        if self.use_dropout:
            self.dropout = nn.Dropout(dropout)
        self._init()

    def _init(self):
        import math
        gain = math.sqrt(1.0 / self.hidden_size)
        for l in range(self.num_layers):
            torch.nn.init.uniform_(self.wx[l], -gain, gain)
            torch.nn.init.uniform_(self.wh[l], -gain, gain)

    def forward(self, inputs: torch.Tensor, prev_state=None, list_next_state: bool = True):
        """inputs [T, B, input_size]; returns (output [T,B,H], next_state)."""
        seq_len, batch_size = inputs.shape[:2]
        prev_state = self._before_forward(inputs, prev_state)
        H, C = prev_state
        x = inputs
        new_h = [None] * self.num_layers
        new_c = [None] * self.num_layers
        for l in range(self.num_layers):
            h, c = H[l], C[l]
            # precompute the input-side GEMM over the whole sequence at once
            gx = torch.matmul(x, self.wx[l])  # [T, B, 4H]
            if self.norm_x is not None:
                gx = self.norm_x[l](gx)
            outputs = []
            # fused HIP lane: LN(gh) + gates + state update in ONE launch per
            # step (vs ~13 eager kernels); see ding/ops/csrc/lstm_ops.hip
            from ding.ops import dispatch as _dispatch
            _fused = (
                self.norm_h is not None and isinstance(x, torch.Tensor) and x.is_cuda
                and x.dtype == torch.float32 and _dispatch.use_hip_autograd(x)
            )
            for t in range(seq_len):
                gh = torch.matmul(h, self.wh[l])
                if _fused:
                    h, c = _dispatch.fused_lstm_cell(
                        gx[t], gh, self.norm_h[l].weight, self.norm_h[l].bias, self.bias[l], c
                    )
                    outputs.append(h)
                    continue
                if self.norm_h is not None:
                    gh = self.norm_h[l](gh)
                gates = gx[t] + gh + self.bias[l]
                i, f, o, u = gates.chunk(4, dim=-1)
                c = torch.sigmoid(f) * c + torch.sigmoid(i) * torch.tanh(u)
                h = torch.sigmoid(o) * torch.tanh(c)
                outputs.append(h)
            x = torch.stack(outputs, dim=0)
            if self.use_dropout and l != self.num_layers - 1:
                x = self.dropout(x)
            new_h[l], new_c[l] = h, c
        next_state = (torch.stack(new_h, dim=0), torch.stack(new_c, dim=0))
        return x, self._after_forward(next_state, list_next_state)


class PytorchLSTM(nn.LSTM, LSTMForwardWrapper):
    """cuDNN/MIOpen-backed LSTM with the same prev_state interface."""

    def forward(self, inputs, prev_state=None, list_next_state: bool = True):
        prev_state = self._before_forward(inputs, prev_state)
        output, next_state = nn.LSTM.forward(self, inputs, prev_state)
        return output, self._after_forward(next_state, list_next_state)


class GRU(nn.GRUCell, LSTMForwardWrapper):
    """GRU over [T, B, C] with the shared state interface (GTrXL option)."""

    def __init__(self, input_size: int, hidden_size: int, num_layers: int = 1):
        super().__init__(input_size, hidden_size)
        self.num_layers = num_layers

    def forward(self, inputs, prev_state=None, list_next_state: bool = True):
        prev_state = self._before_forward(inputs, prev_state)
        h = prev_state[0][0]  # [B, H]
        outputs = []
        for t in range(inputs.shape[0]):
            h = nn.GRUCell.forward(self, inputs[t], h)
            outputs.append(h)
        x = torch.stack(outputs, dim=0)
        next_state = (h.unsqueeze(0), h.unsqueeze(0))
        return x, self._after_forward(next_state, list_next_state)


def get_lstm(
    lstm_type: str,
    input_size: int,
    hidden_size: int,
    num_layers: int = 1,
    norm_type: str = 'LN',
    dropout: float = 0.0,
    seq_len: Optional[int] = None,
    batch_size: Optional[int] = None,
) -> nn.Module:
    assert lstm_type in ('normal', 'pytorch', 'hpc', 'gru')
    if lstm_type in ('normal', 'hpc'):
        # 'hpc' maps to the same module; its pointwise stage dispatches to the
        # HIP kernel lane on GPU once ding/ops ships the fused op.
        return LSTM(input_size, hidden_size, num_layers, norm_type, dropout)
    if lstm_type == 'pytorch':
        return PytorchLSTM(input_size, hidden_size, num_layers)
    return GRU(input_size, hidden_size, num_layers)
