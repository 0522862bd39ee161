"""Middleware executor: per-iteration forward-to-yield / FILO-backward
generator scheduling with optional async mode and distributed labels.

Parity: reference ding/framework/task.py (Task:69, Role:56, forward:260,
backward:284, serial/parallel:308-333, emit/on/wait_for:428-486, the global
``task`` singleton :553).
"""
import asyncio
import concurrent.futures
import enum
import fnmatch
import inspect
import logging
import time
from types import GeneratorType
from typing import Any, Awaitable, Callable, Dict, List, Optional, Set, Union

from .context import Context
from .event_loop import EventLoop

logger = logging.getLogger('ding')


class Role(str, enum.Enum):
    LEARNER = "learner"
    COLLECTOR = "collector"
    EVALUATOR = "evaluator"
    FETCHER = "fetcher"


class VoidMiddleware:

    def __call__(self, ctx):
        return


class Task:
    """Middleware pipeline executor (singleton ``task``).

    Usage::
        with task.start(ctx=OnlineRLContext()):
            task.use(middleware_a)
            task.use(middleware_b)
            task.run(max_step=100)
    """

    role = Role

    def __init__(self):
        self.router = None
        self._initialized = False

    def start(self, async_mode: bool = False, n_async_workers: int = 3, ctx: Optional[Context] = None,
              labels: Optional[Set[str]] = None):
        self.ctx = ctx or Context()
        self._backward_stack: Dict[str, GeneratorType] = {}
        self._roles: Set[str] = set()
        self.labels = labels or set()
        self._middleware: List[Callable] = []
        self._wrappers: List[Callable] = []
        self._finish = False
        self.async_mode = async_mode
        self.n_async_workers = n_async_workers
        self._async_stack: List[Any] = []
        self._thread_pool = None
        self._event_loop = EventLoop(f"task_{id(self)}")
        self._exception = None
        if async_mode:
            self._thread_pool = concurrent.futures.ThreadPoolExecutor(max_workers=n_async_workers)
            self._async_loop = asyncio.new_event_loop()
        else:
            self._async_loop = None

        from .parallel import Parallel
        router = Parallel()
        if router.is_active:
            self.router = router
            self.labels = self.labels | router.labels
            router.on("finish", self._on_remote_finish)

        self._initialized = True
        return self

    # context-manager protocol
    def __enter__(self):
        if not self._initialized:
            self.start()
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        self.stop()

    def stop(self):
        if not self._initialized:
            return
        if self._thread_pool:
            self._thread_pool.shutdown(wait=False)
        if self._async_loop:
            self._async_loop.close()
        self._event_loop.stop()
        self._middleware = []
        self._backward_stack = {}
        self._initialized = False

    # ------------------------------------------------------------- roles
    def add_role(self, role: Role):
        self._roles.add(role)

    def has_role(self, role: Role) -> bool:
        if len(self._roles) == 0:
            return True  # no explicit roles => this node does everything
        return role in self._roles

    @property
    def roles(self) -> Set[str]:
        return self._roles

    @property
    def running(self) -> bool:
        return self._initialized and not self._finish

    # -------------------------------------------------------- composition
    def use(self, fn: Callable, lock: Union[bool, Any] = False) -> 'Task':
        assert callable(fn), f"middleware must be callable: {fn}"
        if lock:
            import threading
            if isinstance(lock, bool):
                lock = threading.Lock()

            def locked_fn(ctx):
                with lock:
                    g = fn(ctx)
                    if isinstance(g, GeneratorType):
                        next(g)  # run eagerly under lock
                return None

            self._middleware.append(self.wrap(locked_fn))
        else:
            self._middleware.append(self.wrap(fn))
        return self

    def use_wrapper(self, fn: Callable) -> 'Task':
        self._wrappers.append(fn)
        # apply retroactively to already-registered middleware
        self._middleware = [fn(m) for m in self._middleware]
        return self

    def wrap(self, fn: Callable) -> Callable:
        for w in self._wrappers:
            fn = w(fn)
        return fn

    def void(self) -> VoidMiddleware:
        return VoidMiddleware()

    # ----------------------------------------------------------- running
    def run(self, max_step: int = int(1e10)) -> None:
        assert self._initialized, "task.start() must be called (use `with task.start():`)"
        if len(self._middleware) == 0:
            return
        for i in range(max_step):
            for fn in self._middleware:
                self.forward(fn)
            # sync point for async mode
            self.sync()
            self.backward()
            self.sync()
            if i == max_step - 1:
                self.finish = True
            if self.finish:
                break
            self.renew()

    def forward(self, fn: Callable, ctx: Optional[Context] = None, async_mode: Optional[bool] = None) -> Any:
        """Run fn(ctx) up to its first yield; stash the generator for the
        backward phase."""
        if ctx is None:
            ctx = self.ctx
        if (async_mode is None and self.async_mode) or async_mode:
            fut = self._thread_pool.submit(self._forward_once, fn, ctx)
            self._async_stack.append(fut)
            return fut
        return self._forward_once(fn, ctx)

    def _forward_once(self, fn: Callable, ctx: Context) -> Any:
        g = fn(ctx)
        if isinstance(g, GeneratorType):
            try:
                next(g)
                key = f"{id(fn)}_{len(self._backward_stack)}"
                self._backward_stack[key] = g
            except StopIteration:
                pass
        return g

    def backward(self, backward_stack: Optional[Dict[str, GeneratorType]] = None) -> None:
        """Resume yielded generators in FILO order."""
        stack = backward_stack if backward_stack is not None else self._backward_stack
        for key in reversed(list(stack.keys())):
            g = stack.pop(key)
            try:
                next(g)
            except StopIteration:
                pass

    def sync(self) -> 'Task':
        if self.async_mode and self._async_stack:
            for fut in self._async_stack:
                exc = fut.exception()
                if exc is not None:
                    raise exc
            self._async_stack = []
        return self

    def renew(self) -> 'Task':
        self.ctx = self.ctx.renew()
        return self

    def serial(self, *fns) -> Callable:
        """Compose several middleware into one sequential middleware."""

        def _serial(ctx):
            stack = {}
            for fn in fns:
                wrapped = self.wrap(fn)
                g = wrapped(ctx)
                if isinstance(g, GeneratorType):
                    try:
                        next(g)
                        stack[f"{id(fn)}_{len(stack)}"] = g
                    except StopIteration:
                        pass
            yield
            self.backward(stack)

        return _serial

    def parallel(self, *fns) -> Callable:
        """Compose middleware running concurrently on the thread pool."""

        def _parallel(ctx):
            if self._thread_pool is None:
                self._thread_pool = concurrent.futures.ThreadPoolExecutor(max_workers=self.n_async_workers)
            futs = [self._thread_pool.submit(self._forward_once, self.wrap(fn), ctx) for fn in fns]
            for fut in futs:
                exc = fut.exception()
                if exc is not None:
                    raise exc

        return _parallel

    # ------------------------------------------------------------ events
    def emit(self, event: str, *args, only_remote: bool = False, only_local: bool = False, **kwargs) -> None:
        if only_remote:
            if self.router:
                self.router.emit(event, *args, **kwargs)
        elif only_local:
            self._event_loop.emit(event, *args, **kwargs)
        else:
            self._event_loop.emit(event, *args, **kwargs)
            if self.router:
                self.router.emit(event, *args, **kwargs)

    def on(self, event: str, fn: Callable) -> None:
        self._event_loop.on(event, fn)
        if self.router:
            self.router.on(event, lambda *a, **k: self._event_loop.emit(event, *a, **k)) \
                if not self.router.listened(event) else None

    def once(self, event: str, fn: Callable) -> None:
        self._event_loop.once(event, fn)

    def off(self, event: str, fn: Optional[Callable] = None) -> None:
        self._event_loop.off(event, fn)

    def wait_for(self, event: str, timeout: float = float("inf"), ignore_timeout_exception: bool = True) -> Any:
        received = []

        def _receiver(*args, **kwargs):
            received.append((args, kwargs))

        self.once(event, _receiver)
        start = time.time()
        while not received:
            if time.time() - start > timeout:
                if ignore_timeout_exception:
                    return None
                raise TimeoutError(f"timeout waiting for event {event}")
            time.sleep(0.01)
        return received[0]

    # ------------------------------------------------------------ finish
    @property
    def finish(self) -> bool:
        return self._finish

    @finish.setter
    def finish(self, value: bool):
        self._finish = value
        if value and self.router and self.router.is_active:
            self.router.emit("finish", True)

    def _on_remote_finish(self, value: bool = True):
        self._finish = True

    def get_attch_to_len(self) -> int:
        return 0 if self.router is None else len(self.router.attach_to)


task = Task()
