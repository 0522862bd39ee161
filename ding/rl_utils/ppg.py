"""PPG joint-phase loss. Parity: reference ding/rl_utils/ppg.py."""
from collections import namedtuple
from typing import Tuple

import torch
import torch.nn.functional as F

ppg_data = namedtuple('ppg_data', ['logit_new', 'logit_old', 'action', 'value_new', 'value_old', 'return_', 'weight'])
ppg_joint_loss = namedtuple('ppg_joint_loss', ['auxiliary_loss', 'behavioral_cloning_loss'])


def ppg_joint_error(data: namedtuple, clip_ratio: float = 0.2, use_value_clip: bool = True) -> namedtuple:
    """Aux phase: clipped value regression + KL(old || new) policy distillation."""
    logit_new, logit_old, action, value_new, value_old, return_, weight = data
    if weight is None:
        weight = torch.ones_like(return_)
    if use_value_clip:
        value_clip = value_old + (value_new - value_old).clamp(-clip_ratio, clip_ratio)
        v1 = (return_ - value_new).pow(2)
        v2 = (return_ - value_clip).pow(2)
        auxiliary_loss = 0.5 * (torch.max(v1, v2) * weight).mean()
    else:
        auxiliary_loss = 0.5 * ((return_ - value_new).pow(2) * weight).mean()
    logp_new = torch.log_softmax(logit_new, dim=-1)
    logp_old = torch.log_softmax(logit_old, dim=-1)
    behavioral_cloning_loss = F.kl_div(logp_new, logp_old.exp(), reduction='batchmean')
    return ppg_joint_loss(auxiliary_loss, behavioral_cloning_loss)
