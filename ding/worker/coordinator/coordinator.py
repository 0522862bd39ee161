"""Legacy distributed coordinator: central task scheduler over the HTTP
substrate.

Parity: reference ding/worker/coordinator/coordinator.py:31 (task state
machine, collector/learner task assignment, resource manager). Per SURVEY §7
the coordinator mode is a compatibility shim over ding.interaction — the
primary distributed path is the event-bus Task runtime + RCCL DP.
"""
import threading
import time
import uuid
from queue import Empty, Queue
from typing import Any, Dict, List, Optional

from ding.interaction import Master
from ding.utils import EasyDict, LockContext, LockContextType, build_logger


class ResourceManager:
    """Track registered workers and their busy/free state."""

    def __init__(self):
        self._resources: Dict[str, dict] = {}
        self._lock = LockContext(LockContextType.THREAD_LOCK)

    def register(self, name: str, info: Optional[dict] = None) -> None:
        with self._lock:
            self._resources[name] = {'busy': False, 'info': info or {}}

    def acquire(self, filter_fn=None) -> Optional[str]:
        with self._lock:
            for name, r in self._resources.items():
                if not r['busy'] and (filter_fn is None or filter_fn(name, r)):
                    r['busy'] = True
                    return name
        return None

    def release(self, name: str) -> None:
        with self._lock:
            if name in self._resources:
                self._resources[name]['busy'] = False

    @property
    def free_count(self) -> int:
        with self._lock:
            return sum(1 for r in self._resources.values() if not r['busy'])


class Coordinator:
    """Assign collect/learn tasks to registered slaves round-robin; gather
    results through the master's finish channel."""

    config = dict(
        collector_task_space=2,
        learner_task_space=1,
        collector_task_timeout=30,
        learner_task_timeout=60,
    )

    def __init__(self, cfg: EasyDict = None):
        self._cfg = EasyDict(dict(self.config, **(cfg or {})))
        self._master = Master()
        self._resource = ResourceManager()
        self._task_queue: Queue = Queue()
        self._results: Dict[str, Any] = {}
        self._end = False
        self._logger, _ = build_logger('./log/coordinator', 'coordinator', need_tb=False)
        self._thread: Optional[threading.Thread] = None

    def start(self):
        self._master.start()
        self._thread = threading.Thread(target=self._schedule_loop, daemon=True)
        self._thread.start()
        return self

    @property
    def master(self) -> Master:
        return self._master

    def register_worker(self, name: str, host: str, port: int, info: Optional[dict] = None) -> None:
        self._master.connect_slave(name, host, port)
        self._resource.register(name, info)

    def submit_task(self, task: dict) -> str:
        task_id = uuid.uuid4().hex
        self._task_queue.put((task_id, task))
        return task_id

    def _schedule_loop(self):
        while not self._end:
            try:
                task_id, task = self._task_queue.get(timeout=0.2)
            except Empty:
                continue
            worker = None
            while worker is None and not self._end:
                worker = self._resource.acquire()
                if worker is None:
                    time.sleep(0.05)
            if worker is None:
                break
            try:
                remote_id = self._master.new_task(worker, task)
                result = self._master.wait_task(remote_id, timeout=self._cfg.learner_task_timeout)
                self._results[task_id] = result
            except Exception as e:
                self._results[task_id] = {'status': 'error', 'error': str(e)}
            finally:
                self._resource.release(worker)

    def wait_task(self, task_id: str, timeout: float = 60.0) -> Any:
        start = time.time()
        while time.time() - start < timeout:
            if task_id in self._results:
                return self._results[task_id]
            time.sleep(0.05)
        raise TimeoutError(task_id)

    def close(self):
        self._end = True
        self._master.close()
