"""Timers & watchdog.

Parity: reference ding/utils/time_helper.py (EasyTimer:48, WatchDog:126) and
time_helper_cuda.py. The GPU timer here uses HIP events through the torch.cuda
API (torch.cuda.Event IS hipEvent on ROCm).
"""
import signal
import time
from typing import Callable

import torch


class TimeWrapperTime:

    def start_time(self):
        self._start = time.perf_counter()

    def end_time(self) -> float:
        return time.perf_counter() - self._start

    @classmethod
    def wrapper(cls, fn: Callable) -> Callable:

        def wrap(*args, **kwargs):
            t = time.perf_counter()
            ret = fn(*args, **kwargs)
            return ret, time.perf_counter() - t

        return wrap


class TimeWrapperCuda(TimeWrapperTime):
    """HIP-event based GPU timer (accurate for async device work)."""

    def __init__(self):
        self._e0 = torch.cuda.Event(enable_timing=True)
        self._e1 = torch.cuda.Event(enable_timing=True)

    def start_time(self):
        self._e0.record()

    def end_time(self) -> float:
        self._e1.record()
        self._e1.synchronize()
        return self._e0.elapsed_time(self._e1) / 1000.0  # ms -> s


def build_time_helper(cfg=None, wrapper_type: str = None):
    use_cuda = wrapper_type == "cuda" or (cfg is not None and getattr(cfg, "cuda", False))
    if use_cuda and torch.cuda.is_available():
        return TimeWrapperCuda
    return TimeWrapperTime


class EasyTimer:
    """``with EasyTimer() as t: ...; t.value`` — seconds elapsed.

    cuda=True uses HIP events when a GPU is present.
    """

    def __init__(self, cuda: bool = True):
        if cuda and torch.cuda.is_available():
            self._impl = TimeWrapperCuda()
        else:
            self._impl = TimeWrapperTime()
        self.value = 0.0

    def __enter__(self):
        self.value = 0.0
        self._impl.start_time()
        return self

    def __exit__(self, *exc):
        self.value = self._impl.end_time()


class TimeoutError(RuntimeError):
    pass


class WatchDog:
    """SIGALRM based timeout guard (main thread only).

    Parity: reference time_helper.py:126. ``timeout=0`` disables.
    """

    def __init__(self, timeout: int = 1):
        self._timeout = int(timeout) + 1 if timeout else 0
        self._failed = False

    def start(self):
        if self._timeout:
            signal.signal(signal.SIGALRM, self._event)
            signal.alarm(self._timeout)

    @staticmethod
    def _event(signum, frame):
        raise TimeoutError("watchdog timeout")

    def stop(self):
        if self._timeout:
            signal.alarm(0)
            signal.signal(signal.SIGALRM, signal.SIG_DFL)
