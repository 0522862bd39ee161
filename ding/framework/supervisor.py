"""Generic child process/thread RPC supervisor.

Parity: reference ding/framework/supervisor.py (Supervisor:207,
ChildProcess:54, ChildThread, SendPayload/RecvPayload:24,34).
"""
import enum
import multiprocessing as mp
import queue
import threading
import traceback
import uuid
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional


class ChildType(str, enum.Enum):
    PROCESS = "process"
    THREAD = "thread"


@dataclass
class SendPayload:
    proc_id: int
    req_id: str = field(default_factory=lambda: uuid.uuid4().hex)
    method: Optional[str] = None
    args: List = field(default_factory=list)
    kwargs: Dict = field(default_factory=dict)


@dataclass
class RecvPayload:
    proc_id: int
    req_id: Optional[str] = None
    method: Optional[str] = None
    data: Any = None
    err: Optional[Exception] = None
    extra: Any = None


def _child_process_loop(proc_id, entry_fn, args, kwargs, send_q, recv_q, shm_callback):
    instance = entry_fn(*args, **kwargs)
    while True:
        payload: SendPayload = send_q.get()
        if payload.method == "__shutdown__":
            break
        try:
            ret = getattr(instance, payload.method)(*payload.args, **payload.kwargs)
            recv = RecvPayload(proc_id=proc_id, req_id=payload.req_id, method=payload.method, data=ret)
            if shm_callback is not None:
                shm_callback(recv)
            recv_q.put(recv)
        except Exception as e:
            traceback.print_exc()
            recv_q.put(RecvPayload(proc_id=proc_id, req_id=payload.req_id, method=payload.method, err=e))


class _Child:

    def __init__(self, proc_id: int, entry_fn: Callable, args, kwargs, ctx, shm_callback=None):
        self.proc_id = proc_id
        self.entry_fn = entry_fn
        self.args = args
        self.kwargs = kwargs
        self.ctx = ctx
        self.shm_callback = shm_callback
        self.send_q = None
        self.proc = None

    def start(self, recv_q):
        raise NotImplementedError

    def shutdown(self):
        raise NotImplementedError

    def restart(self, recv_q):
        self.shutdown()
        self.start(recv_q)


class ChildProcess(_Child):

    def start(self, recv_q):
        ctx = mp.get_context(self.ctx or "fork")
        self.send_q = ctx.Queue()
        self.proc = ctx.Process(
            target=_child_process_loop,
            args=(self.proc_id, self.entry_fn, self.args, self.kwargs, self.send_q, recv_q, self.shm_callback),
            daemon=True,
        )
        self.proc.start()

    def shutdown(self, timeout: float = 1.0):
        if self.proc is not None:
            try:
                self.send_q.put(SendPayload(proc_id=self.proc_id, method="__shutdown__"))
                self.proc.join(timeout=timeout)
                if self.proc.is_alive():
                    self.proc.terminate()
            except Exception:
                pass
            self.proc = None


class ChildThread(_Child):

    def start(self, recv_q):
        self.send_q = queue.Queue()
        self._stop = False

        def loop():
            instance = self.entry_fn(*self.args, **self.kwargs)
            while not self._stop:
                try:
                    payload = self.send_q.get(timeout=0.2)
                except queue.Empty:
                    continue
                if payload.method == "__shutdown__":
                    break
                try:
                    ret = getattr(instance, payload.method)(*payload.args, **payload.kwargs)
                    recv_q.put(RecvPayload(self.proc_id, payload.req_id, payload.method, ret))
                except Exception as e:
                    recv_q.put(RecvPayload(self.proc_id, payload.req_id, payload.method, None, e))

        self.proc = threading.Thread(target=loop, daemon=True)
        self.proc.start()

    def shutdown(self, timeout: float = 1.0):
        self._stop = True
        if self.proc is not None:
            self.send_q.put(SendPayload(proc_id=self.proc_id, method="__shutdown__"))
            self.proc.join(timeout=timeout)
            self.proc = None


class Supervisor:

    TYPE_MAPPING = {ChildType.PROCESS: ChildProcess, ChildType.THREAD: ChildThread}

    def __init__(self, type_: ChildType, mp_ctx: Optional[str] = None):
        self._type = type_
        self._mp_ctx = mp_ctx
        self._children: List[_Child] = []
        self._recv_q = None
        self._running = False

    def register(self, entry_fn: Callable, *args, shm_callback: Optional[Callable] = None, **kwargs) -> None:
        proc_id = len(self._children)
        cls = self.TYPE_MAPPING[self._type]
        self._children.append(cls(proc_id, entry_fn, args, kwargs, self._mp_ctx, shm_callback))

    def start_link(self) -> None:
        if self._running:
            return
        if self._type == ChildType.PROCESS:
            ctx = mp.get_context(self._mp_ctx or "fork")
            self._recv_q = ctx.Queue()
        else:
            self._recv_q = queue.Queue()
        for child in self._children:
            child.start(self._recv_q)
        self._running = True

    def send(self, payload: SendPayload) -> None:
        self._children[payload.proc_id].send_q.put(payload)

    def recv(self, ignore_err: bool = False, timeout: Optional[float] = None) -> RecvPayload:
        payload: RecvPayload = self._recv_q.get(timeout=timeout)
        if payload.err is not None and not ignore_err:
            raise payload.err
        return payload

    def recv_all(self, send_payloads: List[SendPayload], ignore_err: bool = False,
                 callback: Optional[Callable] = None, timeout: Optional[float] = None) -> List[RecvPayload]:
        pending = {p.req_id: i for i, p in enumerate(send_payloads)}
        results: List[Optional[RecvPayload]] = [None] * len(send_payloads)
        while pending:
            payload = self.recv(ignore_err=ignore_err, timeout=timeout)
            if payload.req_id in pending:
                results[pending.pop(payload.req_id)] = payload
                if callback is not None:
                    callback(payload)
        return results

    def shutdown(self, timeout: float = 1.0) -> None:
        for child in self._children:
            child.shutdown(timeout=timeout)
        self._running = False

    def __del__(self):
        try:
            self.shutdown(timeout=0.1)
        except Exception:
            pass
