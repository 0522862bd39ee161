"""Importance sampling weights between target and behaviour outputs.

Parity: reference ding/rl_utils/isw.py (compute_importance_weights).
"""
from typing import Union

import torch
from torch.distributions import Categorical, Independent, Normal


def compute_importance_weights(
    target_output: Union[torch.Tensor, dict],
    behaviour_output: Union[torch.Tensor, dict],
    action: torch.Tensor,
    action_space_type: str = 'discrete',
    requires_grad: bool = False,
):
    """IS ratio pi_target(a|s) / pi_behaviour(a|s); logits in, ratio out.

    discrete: target/behaviour_output are logits [T, B, N], action [T, B].
    continuous: dicts with 'mu'/'sigma', action [T, B, D].
    """
    grad_ctx = torch.enable_grad() if requires_grad else torch.no_grad()
    with grad_ctx:
        if action_space_type == 'discrete':
            # explicit log-softmax (Categorical's arg validation hides a
            # device->host sync, which also breaks hipGraph capture)
            lp_t = torch.log_softmax(target_output, -1).gather(-1, action.long().unsqueeze(-1)).squeeze(-1)
            lp_b = torch.log_softmax(behaviour_output, -1).gather(-1, action.long().unsqueeze(-1)).squeeze(-1)
            rhos = lp_t - lp_b
        elif action_space_type == 'continuous':
            dist_t = Independent(Normal(target_output['mu'], target_output['sigma']), 1)
            dist_b = Independent(Normal(behaviour_output['mu'], behaviour_output['sigma']), 1)
            rhos = dist_t.log_prob(action) - dist_b.log_prob(action)
        else:
            raise ValueError(action_space_type)
        return torch.exp(rhos)
