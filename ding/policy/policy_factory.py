"""Random-policy factory for warm-up collection (``random_collect_size``).

Parity: reference ding/policy/policy_factory.py (PolicyFactory:11,
get_random_policy:97). Design difference: instead of switching on gym space
types, the forward function defaults to asking the env (or env manager) for
``random_action()`` — every ding env implements it — so the same factory
covers discrete, continuous, hybrid and masked MARL action spaces without a
gym dependency. A custom ``forward_fn`` overrides this for exotic spaces.
"""
from collections import namedtuple
from typing import Any, Callable, Dict, Optional

import torch

random_collect_function = namedtuple(
    'random_collect_function', ['forward', 'process_transition', 'get_train_sample', 'reset', 'get_attribute']
)


def _space_forward(action_space) -> Callable:
    """Forward sampling from an explicit action space (gym-style .sample())."""

    def forward(data: Dict[int, Any], *args, **kwargs) -> Dict[int, Any]:
        out = {}
        for env_id, obs in data.items():
            if isinstance(action_space, (list, tuple)):  # one space per agent
                if isinstance(obs, dict) and 'action_mask' in obs:
                    mask = torch.as_tensor(obs['action_mask'], dtype=torch.float32)
                    logit = torch.where(mask == 0.0, torch.full_like(mask, -1e8), torch.ones_like(mask))
                    act = torch.distributions.Categorical(logits=logit).sample()
                    out[env_id] = {'action': act, 'logit': logit}
                else:
                    out[env_id] = {'action': torch.as_tensor([s.sample() for s in action_space])}
            elif hasattr(action_space, 'spaces') and isinstance(getattr(action_space, 'spaces'), dict):
                out[env_id] = {'action': torch.as_tensor([s.sample() for s in action_space.spaces.values()])}
            else:
                sample = action_space.sample()
                if isinstance(sample, int):
                    out[env_id] = {'action': torch.LongTensor([sample])}
                else:
                    out[env_id] = {'action': torch.as_tensor(sample)}
        return out

    return forward


class PolicyFactory:
    """Builds collect-mode views with replaced forward functions (random
    exploration before the learner has gradients to act on)."""

    @staticmethod
    def get_random_policy(
        policy: Any,
        action_space: Any = None,
        forward_fn: Optional[Callable] = None,
    ) -> Any:
        assert not (action_space is None and forward_fn is None), \
            "provide an action_space to sample from or a custom forward_fn"
        fwd = forward_fn if forward_fn is not None else _space_forward(action_space)

        def reset(*args, **kwargs) -> None:
            pass

        return random_collect_function(
            fwd, policy.process_transition, policy.get_train_sample, reset, policy.get_attribute
        )


def get_random_policy(cfg, policy, env) -> Any:
    """Entry helper: policies whose transitions embed policy-only data keep
    collecting with themselves; everyone else gets the random view."""
    if cfg.policy.get('transition_with_policy_data', False):
        return policy
    return PolicyFactory.get_random_policy(policy, action_space=env.action_space)
