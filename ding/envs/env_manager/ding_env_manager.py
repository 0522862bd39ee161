"""One-call env-manager setup used by the simplified (bonus/PPOF) API.
Parity: reference ding/envs/env_manager/ding_env_manager.py:7."""
from functools import partial
from typing import Optional

from .base_env_manager import BaseEnvManagerV2
from .subprocess_env_manager import SubprocessEnvManagerV2


def setup_ding_env_manager(env, env_num: int, context: Optional[str] = None, debug: bool = False,
                           caller: str = 'collector') -> BaseEnvManagerV2:
    assert caller in ('evaluator', 'collector')
    if debug:
        env_cls = BaseEnvManagerV2
        manager_cfg = env_cls.default_config()
    else:
        env_cls = SubprocessEnvManagerV2
        manager_cfg = env_cls.default_config()
        if context is not None:
            manager_cfg.context = context
    return env_cls([partial(env.clone, caller) for _ in range(env_num)], manager_cfg)
