"""Comm adapters for the legacy parallel pipeline: HTTP-slave wrappers that
run a learner / collector as a remote worker, exchanging policies and data
through the shared filesystem.

Parity: reference ding/worker/learner/comm/flask_fs_learner.py
(FlaskFileSystemLearner:72, SendPolicyHook:332, SendLearnInfoHook:373) and
collector/comm/flask_fs_collector.py (FlaskFileSystemCollector).
Re-designed around our compact interaction.Slave: task verbs arrive as
{'name': ...} dicts; tensors move as torch.save files under ``path_data`` /
``path_policy``.
"""
import os
import uuid
from typing import Any, List, Union

import torch

from ding.interaction import Slave, TaskFail
from ding.utils import COMM_COLLECTOR_REGISTRY, COMM_LEARNER_REGISTRY, EasyDict


@COMM_LEARNER_REGISTRY.register('flask_fs')
class FlaskFileSystemLearner(Slave):
    """Learner worker: receives {learner_start_task, learner_get_data_task,
    learner_learn_task, learner_close_task}; data referenced by FS paths."""

    def __init__(
        self,
        cfg: EasyDict,
        host: str = '127.0.0.1',
        port: int = 0,
    ):
        super().__init__(host=host, port=port)
        self._cfg = cfg or EasyDict({})
        self._path_data = self._cfg.get('path_data', './data')
        self._path_policy = self._cfg.get('path_policy', './policy')
        os.makedirs(self._path_data, exist_ok=True)
        os.makedirs(self._path_policy, exist_ok=True)
        self._learner = None
        self._policy_id = None
        self._data_demand = []

    def _process_task(self, task: dict) -> Union[dict, 'TaskFail']:
        name = task.get('name')
        if name == 'resource':
            return {'gpu': torch.cuda.device_count(), 'cpu': os.cpu_count()}
        if name == 'learner_start_task':
            task_info = task['task_info']
            self._policy_id = task_info.get('policy_id', 'policy_{}.pth'.format(uuid.uuid4().hex[:6]))
            from ding.worker import BaseLearner
            from ding.policy import create_policy
            policy = create_policy(EasyDict(task_info['policy']), enable_field=['learn'])
            self._learner = BaseLearner(
                EasyDict(task_info.get('learner_cfg', {})), policy.learn_mode,
                exp_name=task_info.get('exp_name', 'comm_learner')
            )
            return {'message': 'learner started'}
        if name == 'learner_get_data_task':
            batch_size = 0
            if self._learner is not None:
                batch_size = self._learner.policy.get_attribute('batch_size')
            demand = {
                'task_id': task.get('task_id'),
                'buffer_id': task.get('buffer_id'),
                'batch_size': batch_size,
            }
            self._data_demand.append(demand)
            return demand
        if name == 'learner_learn_task':
            assert self._learner is not None, "learner_start_task must come first"
            data_meta = task['data']
            data = self._load_data(data_meta)
            self._learner.train(data, task.get('envstep', -1))
            policy_path = os.path.join(self._path_policy, self._policy_id)
            torch.save(self._learner.policy.state_dict(), policy_path)
            return {
                'policy_id': self._policy_id,
                'train_iter': self._learner.train_iter,
                'learner_done': self._learner.train_iter >= task.get('max_train_iter', float('inf')),
                'learn_info': self._learner.learn_info,
            }
        if name == 'learner_close_task':
            if self._learner is not None:
                self._learner.close()
                self._learner = None
            return {'message': 'closed'}
        return TaskFail(result={'message': f'unknown task {name}'})

    def _load_data(self, meta: Union[dict, list]) -> List[Any]:
        """meta: list of file paths (or {'path': ...} dicts) under path_data."""
        out = []
        metas = meta if isinstance(meta, list) else [meta]
        for m in metas:
            path = m['path'] if isinstance(m, dict) else m
            if not os.path.isabs(path):
                path = os.path.join(self._path_data, path)
            loaded = torch.load(path, map_location='cpu', weights_only=False)
            if isinstance(loaded, list):
                out.extend(loaded)
            else:
                out.append(loaded)
        return out


@COMM_COLLECTOR_REGISTRY.register('flask_fs')
class FlaskFileSystemCollector(Slave):
    """Collector worker: receives {collector_start_task, collector_data_task,
    collector_close_task}; writes collected samples under path_data and
    reloads policies from path_policy."""

    def __init__(self, cfg: EasyDict, host: str = '127.0.0.1', port: int = 0):
        super().__init__(host=host, port=port)
        self._cfg = cfg or EasyDict({})
        self._path_data = self._cfg.get('path_data', './data')
        self._path_policy = self._cfg.get('path_policy', './policy')
        os.makedirs(self._path_data, exist_ok=True)
        self._collector = None
        self._policy = None
        self._env = None

    def _process_task(self, task: dict) -> Union[dict, 'TaskFail']:
        name = task.get('name')
        if name == 'resource':
            return {'gpu': torch.cuda.device_count(), 'cpu': os.cpu_count()}
        if name == 'collector_start_task':
            task_info = task['task_info']
            from functools import partial
            from ding.envs import create_env_manager, get_vec_env_setting
            from ding.policy import create_policy
            from ding.worker import create_serial_collector
            cfg = EasyDict(task_info['cfg'])
            env_fn, collector_env_cfg, _ = get_vec_env_setting(cfg.env)
            manager_cfg = EasyDict(dict(cfg.env.get('manager', {'type': 'base'})))
            manager_cfg.type = manager_cfg.get('type', 'base')
            env = create_env_manager(manager_cfg, [partial(env_fn, cfg=c) for c in collector_env_cfg])
            env.seed(cfg.get('seed', 0))
            self._policy = create_policy(cfg.policy, enable_field=['collect'])
            self._collector = create_serial_collector(
                cfg.policy.collect.get('collector', EasyDict({'type': 'sample'})), env=env,
                policy=self._policy.collect_mode, exp_name=cfg.get('exp_name', 'comm_collector')
            )
            return {'message': 'collector started'}
        if name == 'collector_data_task':
            assert self._collector is not None
            policy_path = task.get('policy_path')
            if policy_path:
                if not os.path.isabs(policy_path):
                    policy_path = os.path.join(self._path_policy, policy_path)
                if os.path.exists(policy_path):
                    sd = torch.load(policy_path, map_location='cpu', weights_only=False)
                    self._policy.collect_mode.load_state_dict(sd)
            n_sample = task.get('n_sample', None)
            data = self._collector.collect(n_sample=n_sample, policy_kwargs=task.get('policy_kwargs'))
            fname = 'data_{}.pth'.format(uuid.uuid4().hex[:8])
            torch.save(data, os.path.join(self._path_data, fname))
            return {
                'data_path': fname,
                'sample_count': len(data),
                'envstep': self._collector.envstep,
                'collector_done': False,
            }
        if name == 'collector_close_task':
            if self._collector is not None:
                self._collector.close()
                self._collector = None
            return {'message': 'closed'}
        return TaskFail(result={'message': f'unknown task {name}'})
