"""Parallel (event-bus) training pipeline: collector nodes + learner node
exchanging context/model over the TCP bus.

Parity: reference ding/entry/parallel_entry.py:14 — re-expressed on the new
Task/Middleware runtime (the reference's gen-1 coordinator mode is legacy;
SURVEY §7 design stance).
"""
import copy
from functools import partial
from typing import Optional, Tuple, Union

from ding.config import compile_config, read_config
from ding.utils import EasyDict, set_pkg_seed


def parallel_pipeline(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    n_parallel_workers: int = 2,
    topology: str = 'mesh',
) -> None:
    """Spawn 1 learner + (n-1) collectors over the event bus."""
    from ding.framework.parallel import Parallel
    if isinstance(input_cfg, str):
        cfg, create_cfg = read_config(input_cfg)
    else:
        cfg, create_cfg = copy.deepcopy(input_cfg[0]), copy.deepcopy(input_cfg[1])
    Parallel.runner(
        n_parallel_workers=n_parallel_workers, topology=topology
    )(partial(_parallel_main, cfg=cfg, create_cfg=create_cfg, seed=seed))


def _parallel_main(cfg: EasyDict, create_cfg: EasyDict, seed: int) -> None:
    from ding.envs import BaseEnvManagerV2, get_vec_env_setting
    from ding.data import DequeBuffer
    from ding.framework import task, OnlineRLContext, Parallel
    from ding.framework.middleware import (
        ContextExchanger, ModelExchanger, OffPolicyLearner, StepCollector, data_pusher, eps_greedy_handler,
        interaction_evaluator, termination_checker, CkptSaver,
    )
    from ding.policy import create_policy
    from ding.config import compile_config

    router = Parallel()
    cfg = compile_config(cfg, seed=seed, auto=True, create_cfg=create_cfg, save_cfg=router.node_id == 0)
    env_fn, collector_env_cfg, evaluator_env_cfg = get_vec_env_setting(cfg.env)
    set_pkg_seed(seed + (router.node_id or 0), use_cuda=cfg.policy.cuda)
    policy = create_policy(cfg.policy, enable_field=['learn', 'collect', 'eval'])

    with task.start(ctx=OnlineRLContext()):
        if router.node_id == 0:
            task.add_role(task.role.LEARNER)
        else:
            task.add_role(task.role.COLLECTOR)

        task.use(ContextExchanger(skip_n_iter=1))
        task.use(ModelExchanger(policy.get_attribute('model')))

        if task.has_role(task.role.COLLECTOR):
            collector_env = BaseEnvManagerV2(
                env_fn=[partial(env_fn, cfg=c) for c in collector_env_cfg], cfg=cfg.env.manager
            )
            collector_env.seed(seed + router.node_id)
            task.use(eps_greedy_handler(cfg))
            task.use(StepCollector(cfg, policy.collect_mode, collector_env))
        if task.has_role(task.role.LEARNER):
            evaluator_env = BaseEnvManagerV2(
                env_fn=[partial(env_fn, cfg=c) for c in evaluator_env_cfg], cfg=cfg.env.manager
            )
            evaluator_env.seed(seed, dynamic_seed=False)
            buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)
            task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
            task.use(data_pusher(cfg, buffer_))
            task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
            task.use(CkptSaver(policy, cfg.exp_name, train_freq=1000))
        task.use(termination_checker(max_env_step=cfg.get('max_env_step', int(1e10))))
        task.run()
