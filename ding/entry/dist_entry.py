"""Legacy distributed-mode launchers: one process per role (coordinator /
learner / collector), tasks flowing commander -> workers over the comm
adapters, data/policies over the shared filesystem.

Parity: reference ding/entry/dist_entry.py (dist_prepare_config:16,
dist_launch_coordinator:49, dist_launch_learner:92,
dist_launch_collector:116). Re-designed around our compact
interaction/comm stack: the coordinator drives a parallel commander's task
loop directly against in-process (or HTTP-attached) comm workers; the
modern path for new code remains the event-bus Task runtime (`ditask`).
"""
import time
from typing import Any, Dict, Optional

from ding.utils import EasyDict
from ding.worker.comm import FlaskFileSystemCollector, FlaskFileSystemLearner
from ding.worker.coordinator.parallel_commander import create_parallel_commander


def dist_prepare_config(cfg: EasyDict, seed: int = 0, platform: Optional[str] = None, **kwargs) -> EasyDict:
    """Fill in per-role addresses/ports (optionally from SLURM/k8s env)."""
    cfg = EasyDict(cfg)
    if platform:
        from ding.entry.cli_parsers import PLATFORM_PARSERS
        task = PLATFORM_PARSERS[platform](**kwargs)
        cfg.system = EasyDict(dict(cfg.get('system', {}), **task))
    cfg.seed = seed
    return cfg


def dist_launch_learner(cfg: EasyDict, host: str = '127.0.0.1', port: int = 0) -> FlaskFileSystemLearner:
    """Start a learner worker (HTTP slave) and return it."""
    worker = FlaskFileSystemLearner(EasyDict(cfg.get('comm', {})), host=host, port=port)
    worker.start()
    return worker


def dist_launch_collector(cfg: EasyDict, host: str = '127.0.0.1', port: int = 0) -> FlaskFileSystemCollector:
    worker = FlaskFileSystemCollector(EasyDict(cfg.get('comm', {})), host=host, port=port)
    worker.start()
    return worker


def dist_launch_coordinator(
    cfg: EasyDict,
    learner: Optional[FlaskFileSystemLearner] = None,
    collector: Optional[FlaskFileSystemCollector] = None,
    max_cycles: int = 10,
) -> Dict[str, Any]:
    """Drive the commander loop: spawn collector tasks, feed collected data
    paths to the learner, stop when the commander (or cycle budget) says so.

    For in-process testing the workers can be passed directly; over HTTP the
    same dicts travel through interaction.Master task posts.
    """
    commander = create_parallel_commander(EasyDict(cfg.commander))
    history = {'collect': [], 'learn': []}

    def _run(worker, task: dict):
        out = worker._process_task(task)
        from ding.interaction import TaskFail
        if isinstance(out, TaskFail):
            raise RuntimeError(f"task failed: {out.result}")
        return out

    started = {'learner': False, 'collector': False}
    policy_id = None
    learner_task_id = None
    for _ in range(max_cycles):
        l_task = commander.get_learner_task()
        if l_task is not None and learner is not None and not started['learner']:
            policy_id = l_task.get('policy_id', 'policy.pth')
            _run(learner, {
                'name': 'learner_start_task',
                'task_info': {'policy': l_task['policy'], 'learner_cfg': l_task.get('learner_cfg', {}),
                              'policy_id': policy_id},
            })
            started['learner'] = True
            learner_task_id = l_task['task_id']
        c_task = commander.get_collector_task()
        if c_task is None:
            time.sleep(0.01)
            continue
        if collector is not None and not started['collector']:
            _run(collector, {'name': 'collector_start_task', 'task_info': {'cfg': c_task['collector_cfg'].cfg}})
            started['collector'] = True
        out = _run(collector, {
            'name': 'collector_data_task',
            'n_sample': c_task['collector_cfg'].get('n_sample', None),
            'policy_path': policy_id if started['learner'] else None,
            'policy_kwargs': c_task['collector_cfg'].get('collect_setting', None),
        })
        history['collect'].append(out)
        stop = commander.finish_collector_task(c_task['task_id'], out)
        if started['learner']:
            lout = _run(learner, {'name': 'learner_learn_task', 'data': [out['data_path']],
                                  'envstep': out['envstep']})
            history['learn'].append(lout)
            commander.update_learner_info(learner_task_id, lout)
        if stop:
            break
    if learner is not None and started['learner']:
        _run(learner, {'name': 'learner_close_task'})
    if collector is not None and started['collector']:
        _run(collector, {'name': 'collector_close_task'})
    return history


def _operator_server_from_cfg(cfg: EasyDict):
    """Build the DI-orchestrator client from cfg/env (reference
    dist_entry.py:249-340 k8s replica commands)."""
    from ding.utils import OperatorServer, get_operator_server_kwargs
    kwargs = get_operator_server_kwargs(cfg.get('system', EasyDict({})))
    server = OperatorServer(**kwargs)
    server.set_worker_type('coordinator')
    return server


def dist_add_replicas(cfg: EasyDict, n_collectors: int = 0, n_learners: int = 0):
    """Ask the orchestrator to scale UP collector/learner replicas."""
    server = _operator_server_from_cfg(cfg)
    ok, code, msg, data = server.post_replicas({'collectors': n_collectors, 'learners': n_learners})
    if not ok:
        raise RuntimeError(f'add replicas failed ({code}): {msg}')
    return data


def dist_delete_replicas(cfg: EasyDict, n_collectors: int = 0, n_learners: int = 0):
    """Ask the orchestrator to scale DOWN collector/learner replicas."""
    server = _operator_server_from_cfg(cfg)
    ok, code, msg, data = server.delete_replicas(n_collectors, n_learners)
    if not ok:
        raise RuntimeError(f'delete replicas failed ({code}): {msg}')
    return data


def dist_restart_replicas(cfg: EasyDict, collectors=None, learners=None):
    """Report failed replicas so the orchestrator restarts them."""
    server = _operator_server_from_cfg(cfg)
    ok, code, msg, data = server.post_replicas_failed(collectors or [], learners or [])
    if not ok:
        raise RuntimeError(f'restart replicas failed ({code}): {msg}')
    return data
