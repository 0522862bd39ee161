"""Shared policy helpers.

Parity: reference ding/policy/common_utils.py (default_preprocess_learn:28,
single_env_forward_wrapper:101).
"""
from typing import Any, Callable, Dict, List

import torch

from ding.utils.data import default_collate
from ding.torch_utils import to_tensor


def default_preprocess_learn(
    data: List[Any],
    use_priority_IS_weight: bool = False,
    use_priority: bool = False,
    use_nstep: bool = False,
    ignore_done: bool = False,
) -> Dict[str, torch.Tensor]:
    """Collate a list of transition dicts into a train batch; normalize
    reward layout for n-step ([B, T] -> [T, B]) and attach IS weights.

    A dict input is treated as already collated (the MI355X middleware keeps
    batches resident as stacked tensors instead of re-collating lists)."""
    if not isinstance(data, dict):
        data = default_collate(data)
    for k in ('obs', 'next_obs'):
        # frame-stack envs keep uint8 frames in the buffer (4x less HBM);
        # models expect float at entry. nstep>1 rewrites next_obs from the
        # (already float) obs stream, nstep==1 keeps the raw env frame — so
        # cast here, once per batch.
        if k in data and isinstance(data[k], torch.Tensor) and data[k].dtype == torch.uint8:
            data[k] = data[k].float()
    if 'value_gamma' in data and isinstance(data['value_gamma'], list):
        data['value_gamma'] = torch.as_tensor(data['value_gamma'], dtype=torch.float32)
    if ignore_done:
        data['done'] = torch.zeros_like(data['done']).float()
    else:
        data['done'] = data['done'].float()

    if use_priority_IS_weight:
        assert use_priority, "priority_IS_weight requires priority"
        if 'priority_IS' in data:
            data['weight'] = data['priority_IS']
        elif 'IS' in data:
            data['weight'] = data['IS']
        else:  # buffer without IS tracking (e.g. naive): uniform weights
            data['weight'] = None
    else:
        data['weight'] = data.get('weight', None)

    if use_nstep:
        # reward collated as [B, nstep] -> [nstep, B]
        if data['reward'].dim() == 1:
            data['reward'] = data['reward'].unsqueeze(1)
        data['reward'] = data['reward'].permute(1, 0).contiguous()
    else:
        if 'reward' in data and isinstance(data['reward'], torch.Tensor) and data['reward'].dim() > 1 \
                and data['reward'].shape[-1] == 1:
            data['reward'] = data['reward'].squeeze(-1)
    if 'action' in data and isinstance(data['action'], torch.Tensor) and data['action'].dim() > 1 \
            and data['action'].shape[-1] == 1 and data['action'].dtype in (torch.int64, torch.int32):
        data['action'] = data['action'].squeeze(-1)
    return data


def single_env_forward_wrapper(forward_fn: Callable) -> Callable:
    """Adapt a batch policy-forward into a single-obs -> single-action fn
    (deploy mode)."""

    def _forward(obs):
        obs = {0: to_tensor(obs, dtype=torch.float32).unsqueeze(0)}
        output = forward_fn(obs)
        action = output[0]['action']
        return action.squeeze(0).detach().cpu().numpy()

    return _forward


def single_env_forward_wrapper_ttorch(forward_fn: Callable, cuda: bool = True) -> Callable:

    def _forward(obs):
        obs = to_tensor(obs, dtype=torch.float32).unsqueeze(0)
        if cuda and torch.cuda.is_available():
            obs = obs.cuda()
        output = forward_fn(obs)
        action = output['action'].squeeze(0).detach().cpu().numpy()
        return action

    return _forward
