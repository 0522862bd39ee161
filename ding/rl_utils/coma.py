"""COMA loss (counterfactual multi-agent PG). Parity: ding/rl_utils/coma.py:10."""
from collections import namedtuple

import torch
import torch.nn.functional as F

from .td import generalized_lambda_returns

coma_data = namedtuple('coma_data', ['logit', 'action', 'q_value', 'target_q_value', 'reward', 'weight'])
coma_loss = namedtuple('coma_loss', ['policy_loss', 'q_value_loss', 'entropy_loss'])


def coma_error(data: namedtuple, gamma: float, lambda_: float) -> namedtuple:
    """logit/q_value/target_q_value: [T,B,A,N]; action [T,B,A]; reward [T,B].

    Critic: TD(lambda) on Q(s, a_taken); actor: counterfactual advantage
    A = Q(a_taken) - sum_a pi(a) Q(a).
    """
    logit, action, q_value, target_q_value, reward, weight = data
    if weight is None:
        weight = torch.ones_like(action)
    q_taken = q_value.gather(-1, action.unsqueeze(-1)).squeeze(-1)  # [T,B,A]
    target_q_taken = target_q_value.gather(-1, action.unsqueeze(-1)).squeeze(-1)
    T, B, A = target_q_taken.shape
    reward_r = reward.unsqueeze(-1).expand_as(target_q_taken).reshape(T, -1)
    target_flat = target_q_taken.reshape(T, -1)
    return_ = generalized_lambda_returns(target_flat, reward_r[:-1], gamma, lambda_)
    return_ = return_.reshape(T - 1, B, A)
    q_value_loss = (F.mse_loss(return_, q_taken[:-1], reduction='none') * weight[:-1]).mean()

    dist = torch.distributions.Categorical(logits=logit)
    logp = dist.log_prob(action)
    baseline = (torch.softmax(logit, dim=-1) * q_value).sum(-1).detach()
    adv = (q_taken - baseline).detach()
    entropy_loss = (dist.entropy() * weight).mean()
    policy_loss = -(logp * adv * weight).mean()
    return coma_loss(policy_loss, q_value_loss, entropy_loss)
