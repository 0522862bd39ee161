"""Local pub/sub event loop with a small thread pool.

Parity: reference ding/framework/event_loop.py (EventLoop:9).
"""
import logging
from collections import defaultdict
from concurrent.futures import ThreadPoolExecutor
from typing import Callable, Optional

logger = logging.getLogger('ding')


class EventLoop:

    loops = {}

    def __init__(self, name: str = "default"):
        self._name = name
        self._listeners = defaultdict(list)
        self._thread_pool = ThreadPoolExecutor(max_workers=2)
        self._exception: Optional[Exception] = None
        self._active = True

    def on(self, event: str, fn: Callable) -> None:
        self._listeners[event].append(fn)

    def off(self, event: str, fn: Optional[Callable] = None) -> None:
        if fn is None:
            self._listeners[event] = []
        else:
            self._listeners[event] = [f for f in self._listeners[event] if f is not fn]

    def once(self, event: str, fn: Callable) -> None:

        def once_fn(*args, **kwargs):
            self.off(event, once_fn)
            fn(*args, **kwargs)

        self.on(event, once_fn)

    def emit(self, event: str, *args, **kwargs) -> None:
        if self._exception:
            raise self._exception
        if self._active:
            self._thread_pool.submit(self._trigger, event, *args, **kwargs)

    def _trigger(self, event: str, *args, **kwargs) -> None:
        for fn in list(self._listeners.get(event, [])):
            try:
                fn(*args, **kwargs)
            except Exception as e:
                self._exception = e
                logger.exception(f"event '{event}' listener raised")

    def listened(self, event: str) -> bool:
        return len(self._listeners.get(event, [])) > 0

    def stop(self) -> None:
        self._active = False
        self._listeners = defaultdict(list)
        self._thread_pool.shutdown(wait=False)
        if self._name in EventLoop.loops:
            del EventLoop.loops[self._name]

    @classmethod
    def get_event_loop(cls, name: str = "default") -> "EventLoop":
        if name not in cls.loops:
            cls.loops[name] = cls(name)
        return cls.loops[name]
