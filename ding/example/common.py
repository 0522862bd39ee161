"""Shared plumbing for the example mains: env builders + config helpers."""
from typing import Callable, Tuple

from ding.envs import BaseEnvManagerV2
from ding.utils import EasyDict, deep_merge_dicts


def cartpole_envs(cfg, collector_n: int = 4, evaluator_n: int = 4) -> Tuple:
    from dizoo.classic_control.cartpole.envs.cartpole_env import CartPoleEnv
    ce = BaseEnvManagerV2(env_fn=[lambda: CartPoleEnv({}) for _ in range(collector_n)], cfg=cfg.env.manager)
    ee = BaseEnvManagerV2(env_fn=[lambda: CartPoleEnv({}) for _ in range(evaluator_n)], cfg=cfg.env.manager)
    ce.seed(0)
    ee.seed(0, dynamic_seed=False)
    return ce, ee


def pendulum_envs(cfg, collector_n: int = 4, evaluator_n: int = 4) -> Tuple:
    from dizoo.classic_control.pendulum.envs.pendulum_env import PendulumEnv
    ce = BaseEnvManagerV2(env_fn=[lambda: PendulumEnv({'act_scale': True}) for _ in range(collector_n)],
                          cfg=cfg.env.manager)
    ee = BaseEnvManagerV2(env_fn=[lambda: PendulumEnv({'act_scale': True}) for _ in range(evaluator_n)],
                          cfg=cfg.env.manager)
    ce.seed(0)
    ee.seed(0, dynamic_seed=False)
    return ce, ee


def compile(main_cfg: dict, create_cfg: dict, exp_name: str) -> EasyDict:
    from ding.config import compile_config
    cfg = compile_config(EasyDict(main_cfg), create_cfg=EasyDict(create_cfg), auto=True, save_cfg=False, seed=0)
    cfg.exp_name = exp_name
    return cfg


def lunarlander_envs(cfg, collector_n: int = 4, evaluator_n: int = 4, continuous: bool = False) -> Tuple:
    from dizoo.box2d.lunarlander.envs.lunarlander_env import LunarLanderEnv
    env_cfg = {'continuous': continuous}
    ce = BaseEnvManagerV2(env_fn=[lambda: LunarLanderEnv(dict(env_cfg)) for _ in range(collector_n)],
                          cfg=cfg.env.manager)
    ee = BaseEnvManagerV2(env_fn=[lambda: LunarLanderEnv(dict(env_cfg)) for _ in range(evaluator_n)],
                          cfg=cfg.env.manager)
    ce.seed(0)
    ee.seed(0, dynamic_seed=False)
    return ce, ee


def offpolicy_main(config_module: str, policy_cls, envs_fn: Callable = None, max_step: int = 1000,
                   exp_name: str = None, use_nstep: bool = False, use_eps: bool = True):
    """Generic off-policy middleware main: evaluator -> (eps) -> collector ->
    (nstep) -> buffer -> learner -> ckpt. The per-algorithm example scripts
    are thin wrappers naming their config + policy (reference ding/example/*)."""
    import importlib

    from ding.data import DequeBuffer
    from ding.framework import OnlineRLContext, task
    from ding.framework.middleware import (
        CkptSaver, OffPolicyLearner, StepCollector, data_pusher, eps_greedy_handler, interaction_evaluator,
        nstep_reward_enhancer, termination_checker,
    )
    mod = importlib.import_module(config_module)
    cfg = compile(mod.main_config, mod.create_config, exp_name or 'exp/example_' + policy_cls.__name__.lower())
    envs_fn = envs_fn or cartpole_envs
    collector_env, evaluator_env = envs_fn(cfg)
    policy = policy_cls(deep_merge_dicts(policy_cls.default_config(), cfg.policy))
    buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
        if use_eps:
            task.use(eps_greedy_handler(cfg))
        task.use(StepCollector(cfg, policy.collect_mode, collector_env,
                               random_collect_size=cfg.policy.get('random_collect_size', 0)))
        if use_nstep:
            task.use(nstep_reward_enhancer(cfg))
        task.use(data_pusher(cfg, buffer_))
        task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
        task.use(CkptSaver(policy, cfg.exp_name, train_freq=1000))
        task.use(termination_checker(max_env_step=int(1e5)))
        task.run(max_step=max_step)
    collector_env.close()
    evaluator_env.close()
    return policy
