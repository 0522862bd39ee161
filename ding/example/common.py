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
