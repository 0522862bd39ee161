"""Offline training pipeline.

Parity: reference ding/entry/serial_entry_offline.py: dataset -> dataloader
-> learner epochs with periodic evaluation.
"""
from functools import partial
from typing import Any, List, Optional, Tuple, Union

import torch
from torch.utils.data import DataLoader

from ding.config import compile_config, read_config
from ding.envs import create_env_manager, get_vec_env_setting
from ding.policy import create_policy
from ding.utils import EasyDict, set_pkg_seed
from ding.utils.data import create_dataset
from ding.worker import BaseLearner, InteractionSerialEvaluator


def serial_pipeline_offline(
    input_cfg: Union[str, Tuple[dict, dict]],
    seed: int = 0,
    env_setting: Optional[List[Any]] = None,
    model: Optional[torch.nn.Module] = None,
    max_train_iter: int = int(1e10),
) -> 'Policy':  # noqa
    if isinstance(input_cfg, str):
        cfg, create_cfg = read_config(input_cfg)
    else:
        import copy
        cfg, create_cfg = copy.deepcopy(input_cfg[0]), copy.deepcopy(input_cfg[1])
    if not create_cfg.policy.type.endswith('_command'):
        create_cfg.policy.type = create_cfg.policy.type + '_command'
    cfg = compile_config(cfg, seed=seed, auto=True, create_cfg=create_cfg, save_cfg=True)

    dataset = create_dataset(cfg)

    def _collate(batch):
        return list(batch)

    dataloader = DataLoader(
        dataset, batch_size=cfg.policy.learn.batch_size, shuffle=True, collate_fn=_collate,
        num_workers=0
    )
    # env for evaluation only
    evaluator = None
    if 'type' in cfg.env.get('manager', {}) or env_setting is not None or 'type' in cfg.env:
        try:
            if env_setting is None:
                env_fn, _, evaluator_env_cfg = get_vec_env_setting(cfg.env, collect=False)
            else:
                env_fn, _, evaluator_env_cfg = env_setting
            manager_cfg = EasyDict(dict(cfg.env.manager))
            manager_cfg.type = cfg.env.manager.get('type', 'base')
            evaluator_env = create_env_manager(manager_cfg, [partial(env_fn, cfg=c) for c in evaluator_env_cfg])
            evaluator_env.seed(cfg.seed, dynamic_seed=False)
        except KeyError:
            evaluator_env = None
    set_pkg_seed(cfg.seed, use_cuda=cfg.policy.cuda)
    policy = create_policy(cfg.policy, model=model)
    learner = BaseLearner(cfg.policy.learn.get('learner', EasyDict({})), policy.learn_mode, exp_name=cfg.exp_name)
    if evaluator_env is not None:
        evaluator = InteractionSerialEvaluator(
            cfg.policy.eval.evaluator, evaluator_env, policy.eval_mode, exp_name=cfg.exp_name
        )

    learner.call_hook('before_run')
    stop = False
    while not stop:
        for train_data in dataloader:
            learner.train(train_data)
            if evaluator is not None and evaluator.should_eval(learner.train_iter):
                stop, _ = evaluator.eval(learner.save_checkpoint, learner.train_iter)
            if learner.train_iter >= max_train_iter:
                stop = True
            if stop:
                break
    learner.call_hook('after_run')
    if evaluator is not None:
        evaluator.close()
    learner.close()
    return policy
