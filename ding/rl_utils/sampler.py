"""Action samplers on raw network outputs.

Parity: reference ding/rl_utils/sampler.py (PPOF collect/eval path). Hybrid
samplers take a dict logit {'action_type': logits, 'action_args': {'mu','sigma'}}.
"""
import torch


class ArgmaxSampler:

    def __call__(self, logit: torch.Tensor) -> torch.Tensor:
        return logit.argmax(dim=-1)


class MultinomialSampler:

    def __call__(self, logit: torch.Tensor) -> torch.Tensor:
        return torch.distributions.Categorical(logits=logit).sample()


class MuSampler:

    def __call__(self, logit) -> torch.Tensor:
        return logit['mu'] if isinstance(logit, dict) else logit.mu


class ReparameterizationSampler:

    def __call__(self, logit) -> torch.Tensor:
        mu = logit['mu'] if isinstance(logit, dict) else logit.mu
        sigma = logit['sigma'] if isinstance(logit, dict) else logit.sigma
        return torch.distributions.Normal(mu, sigma).rsample()


class HybridStochasticSampler:

    def __call__(self, logit):
        action_type = torch.distributions.Categorical(logits=logit['action_type']).sample()
        args = logit['action_args']
        action_args = torch.distributions.Normal(args['mu'], args['sigma']).rsample()
        return {'action_type': action_type, 'action_args': action_args}


class HybridDeterminsticSampler:

    def __call__(self, logit):
        action_type = logit['action_type'].argmax(dim=-1)
        action_args = logit['action_args']['mu']
        return {'action_type': action_type, 'action_args': action_args}
