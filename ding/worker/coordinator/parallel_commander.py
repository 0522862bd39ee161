"""Parallel commanders for the legacy distributed pipeline: task factories
that the Coordinator polls to spawn collector/learner jobs.

Parity: reference ding/worker/coordinator/base_parallel_commander.py
(BaseCommander:9, NaiveCommander:41), solo_parallel_commander.py
(SoloCommander:11), one_vs_one_parallel_commander.py
(OneVsOneCommander:374-equivalent behavior via the league).
"""
import copy
import time
from abc import ABC, abstractmethod
from typing import Optional

from ding.league import create_league
from ding.utils import COMMANDER_REGISTRY, EasyDict, deep_merge_dicts, get_task_uid


class BaseCommander(ABC):

    @abstractmethod
    def get_collector_task(self) -> Optional[dict]:
        raise NotImplementedError

    @abstractmethod
    def get_learner_task(self) -> Optional[dict]:
        raise NotImplementedError

    def judge_collector_finish(self, task_id: str, info: dict) -> bool:
        return bool(info.get('collector_done', False))

    def judge_learner_finish(self, task_id: str, info: dict) -> bool:
        return bool(info.get('learner_done', False))

    def notify_fail_collector_task(self, task: dict) -> None:
        pass

    def notify_fail_learner_task(self, task: dict) -> None:
        pass

    def update_learner_info(self, task_id: str, info: dict) -> None:
        pass


@COMMANDER_REGISTRY.register('naive')
class NaiveCommander(BaseCommander):
    """Spawn exactly ``collector_task_space`` x ``learner_task_space`` tasks
    then stop; bookkeeping only (test/debug commander)."""

    config = dict(type='naive', collector_task_space=1, learner_task_space=1, eval_interval=60)

    def __init__(self, cfg: EasyDict):
        self._cfg = deep_merge_dicts(EasyDict(copy.deepcopy(self.config)), cfg or EasyDict({}))
        self.collector_task_space = self._cfg.collector_task_space
        self.learner_task_space = self._cfg.learner_task_space
        self.collector_task_count = 0
        self.learner_task_count = 0
        self._learner_info = {}
        self._total_collector_env_step = 0

    def get_collector_task(self) -> Optional[dict]:
        if self.collector_task_count < self.collector_task_space:
            self.collector_task_count += 1
            task_id = 'collector_task_{}'.format(get_task_uid())
            return {
                'task_id': task_id,
                'buffer_id': 'test',
                'collector_cfg': self._cfg.get('collector_cfg', EasyDict({})),
                'policy': self._cfg.get('policy', EasyDict({})),
            }
        return None

    def get_learner_task(self) -> Optional[dict]:
        if self.learner_task_count < self.learner_task_space:
            self.learner_task_count += 1
            task_id = 'learner_task_{}'.format(get_task_uid())
            return {
                'task_id': task_id,
                'policy_id': 'test.pth',
                'buffer_id': 'test',
                'learner_cfg': self._cfg.get('learner_cfg', EasyDict({})),
                'policy': self._cfg.get('policy', EasyDict({})),
            }
        return None

    def finish_collector_task(self, task_id: str, finished_task: dict) -> None:
        self._total_collector_env_step += finished_task.get('env_step', 0)

    def finish_learner_task(self, task_id: str, finished_task: dict) -> str:
        return finished_task.get('buffer_id', '')

    def update_learner_info(self, task_id: str, info: dict) -> None:
        self._learner_info[task_id] = info


@COMMANDER_REGISTRY.register('solo')
class SoloCommander(BaseCommander):
    """Single-agent parallel pipeline: one live collector task + one live
    learner task at a time, eps/eval scheduling carried in collector_cfg."""

    config = dict(
        type='solo',
        collector_task_space=2,
        learner_task_space=1,
        eval_interval=60,
    )

    def __init__(self, cfg: EasyDict):
        self._cfg = deep_merge_dicts(EasyDict(copy.deepcopy(self.config)), cfg or EasyDict({}))
        self._collector_task_flags = {}
        self._learner_task_flag = False
        self._last_eval_time = 0.0
        self._learner_info = [{'learner_step': 0}]
        self._total_collector_env_step = 0
        self._evaluator_flag = True

    def get_collector_task(self) -> Optional[dict]:
        if len([k for k, v in self._collector_task_flags.items() if v]) >= self._cfg.collector_task_space:
            return None
        eval_flag = False
        if self._evaluator_flag and (time.time() - self._last_eval_time) > self._cfg.eval_interval:
            eval_flag = True
            self._last_eval_time = time.time()
        task_id = '{}_task_{}'.format('evaluator' if eval_flag else 'collector', get_task_uid())
        self._collector_task_flags[task_id] = True
        collector_cfg = EasyDict(copy.deepcopy(self._cfg.get('collector_cfg', {})))
        collector_cfg.collect_setting = {'eps': self._epsilon()}
        collector_cfg.eval_flag = eval_flag
        return {
            'task_id': task_id,
            'buffer_id': self._cfg.get('buffer_id', 'buffer'),
            'collector_cfg': collector_cfg,
            'policy': self._cfg.get('policy', EasyDict({})),
        }

    def _epsilon(self) -> float:
        # exp decay on learner steps
        step = self._learner_info[-1].get('learner_step', 0)
        start, end, decay = 0.95, 0.1, 10000
        import math
        return end + (start - end) * math.exp(-step / decay)

    def get_learner_task(self) -> Optional[dict]:
        if self._learner_task_flag:
            return None
        self._learner_task_flag = True
        task_id = 'learner_task_{}'.format(get_task_uid())
        return {
            'task_id': task_id,
            'policy_id': self._cfg.get('policy_id', 'policy.pth'),
            'buffer_id': self._cfg.get('buffer_id', 'buffer'),
            'learner_cfg': self._cfg.get('learner_cfg', EasyDict({})),
            'policy': self._cfg.get('policy', EasyDict({})),
        }

    def finish_collector_task(self, task_id: str, finished_task: dict) -> bool:
        self._collector_task_flags[task_id] = False
        self._total_collector_env_step += finished_task.get('env_step', 0)
        if finished_task.get('eval_flag', False):
            return bool(finished_task.get('stop_flag', False))
        return False

    def finish_learner_task(self, task_id: str, finished_task: dict) -> str:
        self._learner_task_flag = False
        return finished_task.get('buffer_id', '')

    def update_learner_info(self, task_id: str, info: dict) -> None:
        self._learner_info.append(info)


@COMMANDER_REGISTRY.register('one_vs_one')
class OneVsOneCommander(BaseCommander):
    """Battle pipeline: league-driven jobs; each collector task carries the
    two players' checkpoints, finish feeds the payoff back."""

    config = dict(
        type='one_vs_one',
        collector_task_space=2,
        learner_task_space=1,
        eval_interval=60,
        league=dict(league_type='one_vs_one'),
    )

    def __init__(self, cfg: EasyDict):
        self._cfg = deep_merge_dicts(EasyDict(copy.deepcopy(self.config)), cfg or EasyDict({}))
        self._league = create_league(self._cfg.league)
        self._collector_task_flags = {}
        self._learner_task_flag = False
        self._learner_info = [{'learner_step': 0}]
        self._last_eval_time = 0.0

    def get_collector_task(self) -> Optional[dict]:
        if len([k for k, v in self._collector_task_flags.items() if v]) >= self._cfg.collector_task_space:
            return None
        eval_flag = (time.time() - self._last_eval_time) > self._cfg.eval_interval
        if eval_flag:
            self._last_eval_time = time.time()
        player_id = self._league.active_players_ids[0]
        job = self._league.get_job_info(player_id, eval_flag)
        task_id = '{}_task_{}'.format('evaluator' if eval_flag else 'collector', get_task_uid())
        self._collector_task_flags[task_id] = True
        collector_cfg = EasyDict(copy.deepcopy(self._cfg.get('collector_cfg', {})))
        collector_cfg.job = job
        collector_cfg.eval_flag = eval_flag
        return {
            'task_id': task_id,
            'buffer_id': self._cfg.get('buffer_id', 'buffer'),
            'collector_cfg': collector_cfg,
            'policy': self._cfg.get('policy', EasyDict({})),
        }

    def get_learner_task(self) -> Optional[dict]:
        if self._learner_task_flag:
            return None
        self._learner_task_flag = True
        return {
            'task_id': 'learner_task_{}'.format(get_task_uid()),
            'policy_id': self._league.active_players_ids[0],
            'buffer_id': self._cfg.get('buffer_id', 'buffer'),
            'learner_cfg': self._cfg.get('learner_cfg', EasyDict({})),
            'policy': self._cfg.get('policy', EasyDict({})),
        }

    def finish_collector_task(self, task_id: str, finished_task: dict) -> bool:
        self._collector_task_flags[task_id] = False
        if 'job_info' in finished_task:
            self._league.finish_job(finished_task['job_info'])
        if finished_task.get('eval_flag', False):
            return bool(finished_task.get('stop_flag', False))
        return False

    def finish_learner_task(self, task_id: str, finished_task: dict) -> str:
        self._learner_task_flag = False
        player_id = self._league.active_players_ids[0]
        self._league.update_active_player({
            'player_id': player_id, 'train_iter': finished_task.get('train_iter', 0)
        })
        self._league.judge_snapshot(player_id)
        return finished_task.get('buffer_id', '')

    def update_learner_info(self, task_id: str, info: dict) -> None:
        self._learner_info.append(info)


def create_parallel_commander(cfg: EasyDict) -> BaseCommander:
    cfg = EasyDict(cfg)
    return COMMANDER_REGISTRY.build(cfg.type, cfg=cfg)
