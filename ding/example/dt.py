"""Offline Decision Transformer on synthetic trajectory tuples (reference
ding/example/dt.py)."""
import torch

from ding.framework import OfflineRLContext, task
from ding.framework.middleware import CkptSaver, offline_data_fetcher, trainer
from ding.policy import create_policy
from ding.utils import EasyDict, deep_merge_dicts


def main(max_step: int = 20, exp_name: str = 'exp/example_dt', n: int = 128):
    from ding.policy.dt import DTPolicy
    K, obs_dim, act_dim = 8, 4, 2
    cfg_p = EasyDict(deep_merge_dicts(DTPolicy.default_config(), EasyDict(dict(
        type='dt', cuda=False,
        model=dict(state_dim=obs_dim, act_dim=act_dim, n_blocks=2, h_dim=32, context_len=K,
                   n_heads=2, drop_p=0.1, continuous=True),
        learn=dict(batch_size=16),
    ))))
    policy = create_policy(cfg_p, enable_field=['learn'])
    # synthetic (timestep, state, action, rtg, mask) tuples
    dataset = [
        (
            torch.arange(K),
            torch.randn(K, obs_dim),
            torch.rand(K, act_dim) * 2 - 1,
            torch.linspace(1, 0, K).unsqueeze(-1),
            torch.ones(K),
        ) for _ in range(n)
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
