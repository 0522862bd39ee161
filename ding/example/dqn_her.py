"""BitFlip DQN + Hindsight Experience Replay middleware (reference
ding/example/dqn_her.py)."""
import torch

from ding.data import DequeBuffer
from ding.envs import BaseEnvManagerV2
from ding.framework import OnlineRLContext, task
from ding.framework.middleware import (
    EpisodeCollector, eps_greedy_handler, her_data_enhancer, interaction_evaluator, termination_checker, trainer,
)
from ding.policy import DQNPolicy
from ding.reward_model import HerRewardModel
from ding.utils import EasyDict
from .common import compile


def main(max_step: int = 200, exp_name: str = 'exp/example_dqn_her', n_bits: int = 6):
    from dizoo.bitflip.envs.bitflip_env import BitFlipEnv
    main_config = EasyDict(dict(
        exp_name=exp_name,
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=1,
                 n_bits=n_bits),
        policy=dict(
            cuda=False, nstep=1, discount_factor=0.98,
            model=dict(obs_shape=2 * n_bits, action_shape=n_bits, encoder_hidden_size_list=[64, 64]),
            learn=dict(update_per_collect=2, batch_size=32, learning_rate=1e-3),
            collect=dict(n_episode=4, unroll_len=1),
            eval=dict(evaluator=dict(eval_freq=100)),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                       replay_buffer=dict(replay_buffer_size=2000)),
        ),
    ))
    create_config = EasyDict(dict(
        env=dict(type='bitflip', import_names=['dizoo.bitflip.envs.bitflip_env']),
        env_manager=dict(type='base'),
        policy=dict(type='dqn'),
    ))
    cfg = compile(main_config, create_config, exp_name)
    ce = BaseEnvManagerV2(env_fn=[lambda: BitFlipEnv({'n_bits': n_bits}) for _ in range(2)], cfg=cfg.env.manager)
    ee = BaseEnvManagerV2(env_fn=[lambda: BitFlipEnv({'n_bits': n_bits}) for _ in range(2)], cfg=cfg.env.manager)
    ce.seed(0)
    ee.seed(0, dynamic_seed=False)
    policy = DQNPolicy(cfg.policy)
    buffer_ = DequeBuffer(size=64)  # stores whole episodes

    def goal_fn(t):
        return t['next_obs'][:n_bits]

    def reward_fn(goal, t):
        ok = torch.allclose(t['next_obs'][:n_bits].float(), torch.as_tensor(goal).float())
        return torch.ones_like(t['reward']) if ok else torch.zeros_like(t['reward'])

    her = HerRewardModel({'her_strategy': 'future', 'her_replay_k': 1,
                          'goal_fn': goal_fn, 'reward_fn': reward_fn})

    def episode_pusher(buffer_):
        def _push(ctx):
            if ctx.episodes is not None:
                for ep in ctx.episodes:
                    buffer_.push(ep)
                ctx.episodes = None
        return _push

    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, ee))
        task.use(eps_greedy_handler(cfg))
        task.use(EpisodeCollector(cfg, policy.collect_mode, ce))
        task.use(episode_pusher(buffer_))
        task.use(her_data_enhancer(cfg, buffer_, her))
        task.use(trainer(cfg, policy.learn_mode))
        task.use(termination_checker(max_env_step=int(1e5)))
        task.run(max_step=max_step)
    ce.close()
    ee.close()
    return policy


if __name__ == '__main__':
    main()
