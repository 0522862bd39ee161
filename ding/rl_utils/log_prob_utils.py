"""Memory-efficient token log-prob extraction for LLM policies.

Parity: reference ding/rl_utils/log_prob_utils.py. ``efficient_method`` avoids
materializing the [B,S,V] log-softmax: logp(a) = logit(a) - logsumexp(logits).
"""
import torch
from torch import Tensor


def naive_method(logits: Tensor, index: Tensor) -> Tensor:
    return torch.log_softmax(logits, dim=-1).gather(-1, index.unsqueeze(-1)).squeeze(-1)


def efficient_method(logits: Tensor, index: Tensor) -> Tensor:
    taken = logits.gather(-1, index.unsqueeze(-1)).squeeze(-1)
    lse = torch.logsumexp(logits, dim=-1)
    return taken - lse


def less_efficient_method(logits: Tensor, index: Tensor) -> Tensor:
    dist = torch.distributions.Categorical(logits=logits)
    return dist.log_prob(index)
