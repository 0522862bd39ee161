"""Offline CQL from a synthetic dataset (reference ding/example/cql.py)."""
import torch

from ding.framework import OfflineRLContext, task
from ding.framework.middleware import CkptSaver, offline_data_fetcher, offline_logger, trainer
from ding.policy import create_policy
from ding.utils import EasyDict, deep_merge_dicts


def main(max_step: int = 20, exp_name: str = 'exp/example_cql', n: int = 256):
    from ding.policy.offline import CQLPolicy
    cfg_p = EasyDict(deep_merge_dicts(CQLPolicy.default_config(), EasyDict(dict(
        type='cql', cuda=False,
        model=dict(obs_shape=3, action_shape=1, twin_critic=True, action_space='reparameterization'),
        learn=dict(batch_size=32, auto_alpha=True),
    ))))
    policy = create_policy(cfg_p, enable_field=['learn'])
    dataset = [
        {
            'obs': torch.randn(3), 'next_obs': torch.randn(3), 'action': torch.rand(1) * 2 - 1,
            'reward': torch.randn(1), 'done': False,
        } for _ in range(n)
    ]
    cfg = EasyDict(dict(exp_name=exp_name, policy=cfg_p))
    with task.start(ctx=OfflineRLContext()):
        task.use(offline_data_fetcher(cfg, dataset))
        task.use(trainer(cfg, policy.learn_mode))
        task.use(CkptSaver(policy, exp_name, train_freq=1000))
        task.run(max_step=max_step)
    return policy


if __name__ == '__main__':
    main()
