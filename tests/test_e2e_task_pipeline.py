"""End-to-end: CartPole DQN through the Task/Middleware pipeline on CPU.

Mirrors the reference's ding/example/dqn.py middleware composition
(SURVEY §3.2 call stack).
"""
import pytest
import torch

from ding.config import compile_config
from ding.envs import BaseEnvManagerV2
from ding.data import DequeBuffer
from ding.policy import DQNPolicy
from ding.framework import task
from ding.framework import OnlineRLContext
from ding.framework.middleware import (
    OffPolicyLearner, StepCollector, interaction_evaluator, data_pusher, eps_greedy_handler, CkptSaver,
    termination_checker, online_logger, nstep_reward_enhancer,
)
from dizoo.classic_control.cartpole.config.cartpole_dqn_config import main_config, create_config
from dizoo.classic_control.cartpole.envs.cartpole_env import CartPoleEnv


def test_cartpole_dqn_pipeline(tmp_path):
    cfg = compile_config(
        main_config, create_cfg=create_config, auto=True, save_cfg=False, seed=0
    )
    cfg.exp_name = str(tmp_path / "cartpole_dqn")
    collector_env = BaseEnvManagerV2(
        env_fn=[lambda: CartPoleEnv({}) for _ in range(4)], cfg=cfg.env.manager
    )
    evaluator_env = BaseEnvManagerV2(
        env_fn=[lambda: CartPoleEnv({}) for _ in range(5)], cfg=cfg.env.manager
    )
    collector_env.seed(0)
    evaluator_env.seed(0, dynamic_seed=False)
    policy = DQNPolicy(cfg.policy)
    buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)

    
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
        task.use(eps_greedy_handler(cfg))
        task.use(StepCollector(cfg, policy.collect_mode, collector_env))
        task.use(data_pusher(cfg, buffer_))
        task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
        task.use(CkptSaver(policy, cfg.exp_name, train_freq=1000))
        task.use(termination_checker(max_env_step=2000))
        task.run(max_step=30)

    assert task.ctx.env_step > 0
    assert task.ctx.train_iter > 0
    collector_env.close()
    evaluator_env.close()


def test_cartpole_ppo_pipeline(tmp_path):
    from ding.policy import PPOPolicy
    from ding.framework.middleware import gae_estimator, multistep_trainer
    from ding.utils import EasyDict, deep_merge_dicts

    ppo_cfg = EasyDict(
        dict(
            exp_name=str(tmp_path / "cartpole_ppo"),
            env=dict(
                collector_env_num=4, evaluator_env_num=2, n_evaluator_episode=2, stop_value=195,
            ),
            policy=dict(
                cuda=False,
                action_space='discrete',
                model=dict(obs_shape=4, action_shape=2, action_space='discrete'),
                learn=dict(epoch_per_collect=2, batch_size=32, learning_rate=3e-4),
                collect=dict(n_sample=64, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
                eval=dict(evaluator=dict(eval_freq=1000, )),
            ),
        )
    )
    cfg = compile_config(ppo_cfg, policy=PPOPolicy, save_cfg=False, seed=0)
    collector_env = BaseEnvManagerV2(env_fn=[lambda: CartPoleEnv({}) for _ in range(4)], cfg=cfg.env.manager)
    evaluator_env = BaseEnvManagerV2(env_fn=[lambda: CartPoleEnv({}) for _ in range(2)], cfg=cfg.env.manager)
    collector_env.seed(0)
    evaluator_env.seed(0, dynamic_seed=False)
    policy = PPOPolicy(cfg.policy)

    
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
        task.use(StepCollector(cfg, policy.collect_mode, collector_env))
        task.use(gae_estimator(cfg, policy.collect_mode))
        task.use(multistep_trainer(policy.learn_mode))
        task.use(termination_checker(max_env_step=1000))
        task.run(max_step=5)

    assert task.ctx.train_iter > 0
    collector_env.close()
    evaluator_env.close()


def test_step_timer_wrapper():
    from ding.framework import StepTimer
    from ding.framework import task as _task

    calls = []

    def mw(ctx):
        calls.append('fwd')
        yield
        calls.append('bwd')

    timer = StepTimer(print_per_step=1)
    wrapped = timer(mw)

    class Ctx:
        total_step = 0

    g = wrapped(Ctx())
    next(g)
    try:
        next(g)
    except StopIteration:
        pass
    assert calls == ['fwd', 'bwd']
    assert len(timer.records['mw']) == 1


@pytest.mark.parametrize('mod', ['dqn', 'dqn_per', 'ppo', 'sac', 'dqn_rnd', 'ppo_offpolicy', 'cql', 'dqn_her', 'dt'])
def test_example_mains(mod, tmp_path):
    """ding/example/* mains run a few pipeline steps end-to-end."""
    import importlib
    m = importlib.import_module(f'ding.example.{mod}')
    m.main(max_step=6, exp_name=str(tmp_path / mod))
