"""Trajectory post-processing: GAE annotation, n-step reward restructuring,
unroll splitting.

Parity: reference ding/rl_utils/adder.py (Adder.get_gae:20,
get_nstep_return_data:97, get_train_sample:158).
"""
import copy
from collections import deque
from typing import Any, Dict, List, Optional

import torch

from ding.utils import lists_to_dicts
from .gae import gae, gae_data


def list_split(data: list, step: int):
    """Split list into ``step``-sized chunks; return (chunks, residual|None)."""
    if len(data) < step:
        return [], data if len(data) > 0 else None
    chunks = [data[i:i + step] for i in range(0, len(data) - step + 1, step)]
    rest = len(chunks) * step
    residual = data[rest:] if rest < len(data) else None
    return chunks, residual


class Adder:

    @classmethod
    def get_gae(
        cls, data: List[Dict[str, Any]], last_value: torch.Tensor, gamma: float, gae_lambda: float, cuda: bool
    ) -> List[Dict[str, Any]]:
        """Annotate each transition dict with 'adv' via a single stacked GAE call."""
        value = torch.stack([d['value'] for d in data])
        next_value = torch.stack([d['value'] for d in data][1:] + [last_value])
        reward = torch.stack([d['reward'] for d in data])
        if reward.dim() == value.dim() + 1 and reward.shape[-1] == 1:
            reward = reward.squeeze(-1)  # [T, 1] env reward vs scalar values
        if cuda:
            value, next_value, reward = value.cuda(), next_value.cuda(), reward.cuda()
        adv = gae(gae_data(value, next_value, reward, None, None), gamma, gae_lambda)
        if cuda:
            adv = adv.cpu()
        for i in range(len(data)):
            data[i]['adv'] = adv[i]
        return data

    @classmethod
    def get_gae_with_default_last_value(
        cls, data: deque, done: bool, gamma: float, gae_lambda: float, cuda: bool
    ) -> List[Dict[str, Any]]:
        if done:
            last_value = torch.zeros_like(data[-1]['value'])
            data = list(data)
        else:
            data = list(data)
            last = data.pop()
            last_value = last['value']
        return cls.get_gae(data, last_value, gamma, gae_lambda, cuda)

    @classmethod
    def get_nstep_return_data(
        cls, data: deque, nstep: int, cum_reward: bool = False, correct_terminate_gamma: bool = True,
        gamma: float = 0.99
    ) -> deque:
        """Rewrite ['next_obs','reward','done'] with n-step values.

        reward becomes the concatenated n-step reward vector ([nstep] per
        sample) unless cum_reward, matching q_nstep_td_error's [T,B] layout
        after collation. Near the trajectory tail, missing steps use zero
        rewards and the final next_obs/done; value_gamma corrects the
        bootstrap discount for truncated tails.
        """
        if nstep == 1:
            return data
        data = list(data)
        L = len(data)
        fake_reward = torch.zeros_like(data[0]['reward'])
        has_next_obs = 'next_obs' in data[0]
        for i in range(L):
            tail = L - i
            if i + nstep < L:
                if has_next_obs:
                    data[i]['next_obs'] = data[i + nstep]['obs']
                if cum_reward:
                    data[i]['reward'] = sum(data[i + j]['reward'] * (gamma ** j) for j in range(nstep))
                else:
                    data[i]['reward'] = torch.cat([data[i + j]['reward'] for j in range(nstep)], dim=-1)
                if correct_terminate_gamma:
                    data[i]['value_gamma'] = gamma ** nstep
            else:
                if has_next_obs:
                    data[i]['next_obs'] = data[-1]['next_obs']
                if cum_reward:
                    data[i]['reward'] = sum(data[i + j]['reward'] * (gamma ** j) for j in range(tail))
                else:
                    data[i]['reward'] = torch.cat(
                        [data[i + j]['reward'] if j < tail else fake_reward for j in range(nstep)], dim=-1
                    )
                data[i]['done'] = data[-1]['done']
                if correct_terminate_gamma:
                    data[i]['value_gamma'] = gamma ** tail
        return deque(data)

    @classmethod
    def get_train_sample(
        cls,
        data: List[Dict[str, Any]],
        unroll_len: int,
        last_fn_type: str = 'last',
        null_transition: Optional[dict] = None,
    ) -> List[Dict[str, Any]]:
        """Split a trajectory into unroll_len slices (RNN training samples).

        last_fn_type: 'drop' | 'last' (borrow from previous slice) |
        'null_padding' (zero transitions with done=True).
        """
        if unroll_len == 1:
            return data
        split_data, residual = list_split(data, step=unroll_len)
        miss_num = 0 if residual is None else unroll_len - len(residual)

        def null_padding():
            template = copy.deepcopy(residual[0])
            template['null'] = True
            if isinstance(template['obs'], dict):
                template['obs'] = {k: torch.zeros_like(v) for k, v in template['obs'].items()}
            else:
                template['obs'] = torch.zeros_like(template['obs'])
            if 'action' in template:
                template['action'] = torch.zeros_like(template['action'])
            template['done'] = True
            template['reward'] = torch.zeros_like(template['reward'])
            if 'value_gamma' in template:
                template['value_gamma'] = 0.0
            return [cls._get_null_transition(template, null_transition) for _ in range(miss_num)]

        if residual is not None:
            if last_fn_type == 'drop':
                pass
            elif last_fn_type == 'last':
                if len(split_data) > 0:
                    borrowed = copy.deepcopy(split_data[-1][-miss_num:])
                    split_data.append(borrowed + residual)
                else:
                    split_data.append(residual + null_padding())
            elif last_fn_type == 'null_padding':
                split_data.append(residual + null_padding())
            else:
                raise ValueError(last_fn_type)
        if len(split_data) > 0:
            split_data = [lists_to_dicts(d, recursive=True) for d in split_data]
        return split_data

    @classmethod
    def _get_null_transition(cls, template: dict, null_transition: Optional[dict] = None) -> dict:
        if null_transition is not None:
            return copy.deepcopy(null_transition)
        return copy.deepcopy(template)


# functional aliases (reference exposes both the class and functions)
get_gae = Adder.get_gae
get_gae_with_default_last_value = Adder.get_gae_with_default_last_value
get_nstep_return_data = Adder.get_nstep_return_data
get_train_sample = Adder.get_train_sample
