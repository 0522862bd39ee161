"""Thread / process / file locks.

Parity: reference ding/utils/lock_helper.py:15-164.
"""
import fcntl
import multiprocessing
import os
import threading
from enum import Enum, unique


@unique
class LockContextType(Enum):
    THREAD_LOCK = 1
    PROCESS_LOCK = 2


_LOCK_TYPE_MAPPING = {
    LockContextType.THREAD_LOCK: threading.Lock,
    LockContextType.PROCESS_LOCK: multiprocessing.Lock,
}


class LockContext:
    """Context-manager lock, thread- or process-scoped."""

    def __init__(self, lock_type: LockContextType = LockContextType.THREAD_LOCK):
        self.lock = _LOCK_TYPE_MAPPING[lock_type]()

    def acquire(self):
        self.lock.acquire()

    def release(self):
        self.lock.release()

    def __enter__(self):
        self.lock.acquire()
        return self

    def __exit__(self, *exc):
        self.lock.release()


rw_lock_mapping = {}


def get_rw_file_lock(name: str, op: str):
    """Named reader/writer thread locks (shared registry)."""
    assert op in ("read", "write")
    if name not in rw_lock_mapping:
        rw_lock_mapping[name] = {"read": threading.Lock(), "write": threading.Lock()}
    return rw_lock_mapping[name][op]


class FcntlContext:
    """Cross-process advisory file lock via fcntl.flock."""

    def __init__(self, lock_path: str):
        self.lock_path = lock_path
        self._fh = None

    def __enter__(self):
        assert self._fh is None
        d = os.path.dirname(self.lock_path)
        if d:
            os.makedirs(d, exist_ok=True)
        self._fh = open(self.lock_path, "w")
        fcntl.flock(self._fh.fileno(), fcntl.LOCK_EX)
        return self

    def __exit__(self, *exc):
        fcntl.flock(self._fh.fileno(), fcntl.LOCK_UN)
        self._fh.close()
        self._fh = None


def get_file_lock(name: str, op: str) -> FcntlContext:
    return FcntlContext(name + ".lock")
