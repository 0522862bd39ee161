"""Process-level event-bus router: spawns N workers, wires a mesh/star
topology over the TCP bus, relays pickled events.

Parity: reference ding/framework/parallel.py (Parallel:23, runner:62,
topology:156-164, listen loop:280, auto-recover:207-236). The reference uses
mpire + pynng; this build uses torch.multiprocessing spawn + the stdlib TCP
bus (ding/framework/message_queue/tcp.py).
"""
import logging
import os
import pickle
import random
import socket as socket_mod
import threading
import time
import traceback
from typing import Any, Callable, Dict, List, Optional, Set

from ding.utils import MQ_REGISTRY
from .event_loop import EventLoop

logger = logging.getLogger('ding')


def _free_ports(n: int, start: int = 15000) -> List[int]:
    ports = []
    p = start + random.randint(0, 2000)
    while len(ports) < n:
        with socket_mod.socket() as s:
            try:
                s.bind(("127.0.0.1", p))
                ports.append(p)
            except OSError:
                pass
        p += 1
    return ports


class Parallel:
    """Per-process singleton router."""

    _instance = None

    def __new__(cls, *args, **kwargs):
        if cls._instance is None:
            cls._instance = super().__new__(cls)
            cls._instance._init_attrs()
        return cls._instance

    def _init_attrs(self):
        self.is_active = False
        self.node_id = None
        self.ngpus = 0
        self.labels: Set[str] = set()
        self.attach_to: List[str] = []
        self._mq = None
        self._event_loop = EventLoop("parallel")
        self._listener_thread = None
        self._retries = 0

    # ------------------------------------------------------------- runner
    @classmethod
    def runner(
        cls,
        n_parallel_workers: int = 1,
        mq_type: str = "tcp",
        protocol: str = "tcp",
        address: str = "127.0.0.1",
        ports: Optional[List[int]] = None,
        topology: str = "mesh",
        labels: Optional[Set[str]] = None,
        attach_to: Optional[List[str]] = None,
        node_ids: Optional[List[int]] = None,
        auto_recover: bool = False,
        max_retries: int = 1,
        startup_interval: float = 0.1,
        redis_host: Optional[str] = None,
        redis_port: Optional[int] = None,
    ) -> Callable:
        """Return a launcher: ``Parallel.runner(...)(main_fn)`` spawns
        n_parallel_workers processes each running main_fn with an active
        router."""
        attach_to = attach_to or []
        assert topology in ("mesh", "star", "alone")

        def _runner(main_fn: Callable, *args, **kwargs):
            if ports is None:
                use_ports = _free_ports(n_parallel_workers)
            elif isinstance(ports, int):
                use_ports = [ports + i for i in range(n_parallel_workers)]
            else:
                use_ports = ports
            nodes = [f"tcp://{address}:{p}" for p in use_ports]
            runner_params = []
            for i in range(n_parallel_workers):
                node_id = node_ids[i] if node_ids else i
                if topology == "mesh":
                    peers = nodes[:i] + attach_to
                elif topology == "star":
                    peers = (nodes[:1] if i != 0 else []) + attach_to
                else:
                    peers = list(attach_to)
                runner_params.append((i, node_id, nodes[i], peers))
            if n_parallel_workers == 1:
                _subprocess_runner(
                    runner_params[0], mq_type, labels, auto_recover, max_retries, main_fn, args, kwargs
                )
            else:
                import multiprocessing as mp
                ctx = mp.get_context("spawn")
                procs = []
                for param in runner_params:
                    p = ctx.Process(
                        target=_subprocess_runner,
                        args=(param, mq_type, labels, auto_recover, max_retries, main_fn, args, kwargs),
                        daemon=False,
                    )
                    p.start()
                    procs.append(p)
                    time.sleep(startup_interval)
                for p in procs:
                    p.join()
                for p in procs:
                    if p.exitcode != 0:
                        raise RuntimeError(f"parallel worker failed with exit code {p.exitcode}")

        return _runner

    def _run(self, node_id: int, listen_to: str, attach_to: List[str], mq_type: str = "tcp",
             labels: Optional[Set[str]] = None):
        self.node_id = node_id
        self.attach_to = attach_to
        self.labels = labels or set()
        import ding.framework.message_queue  # populate MQ_REGISTRY in spawned procs
        self._mq = MQ_REGISTRY.get(mq_type)(listen_to=listen_to, attach_to=attach_to)
        self._mq.listen()
        self.is_active = True
        self._listener_thread = threading.Thread(target=self.listen, daemon=True)
        self._listener_thread.start()

    # ------------------------------------------------------------- events
    def listen(self):
        while self.is_active:
            try:
                topic, payload = self._mq.recv()
            except (ConnectionAbortedError, OSError):
                break
            self._handle_message(topic, payload)

    def _handle_message(self, topic: str, payload: bytes):
        try:
            args, kwargs = pickle.loads(payload)
        except Exception:
            logger.exception("failed to unpickle event payload")
            return
        self._event_loop.emit(topic, *args, **kwargs)

    def on(self, event: str, fn: Callable) -> None:
        self._event_loop.on(event, fn)

    def once(self, event: str, fn: Callable) -> None:
        self._event_loop.once(event, fn)

    def off(self, event: str, fn: Optional[Callable] = None) -> None:
        self._event_loop.off(event, fn)

    def listened(self, event: str) -> bool:
        return self._event_loop.listened(event)

    def emit(self, event: str, *args, **kwargs) -> None:
        if self._mq is None:
            return
        payload = pickle.dumps((args, kwargs), protocol=pickle.HIGHEST_PROTOCOL)
        self._mq.publish(event, payload)

    def get_node_addrs(self) -> List[str]:
        return list(self.attach_to)

    def stop(self):
        self.is_active = False
        if self._mq is not None:
            self._mq.stop()
        self._event_loop.stop()
        self._event_loop = EventLoop("parallel")

    @classmethod
    def reset(cls):
        if cls._instance is not None:
            try:
                cls._instance.stop()
            except Exception:
                pass
        cls._instance = None


def _subprocess_runner(param, mq_type, labels, auto_recover, max_retries, main_fn, args, kwargs):
    """Entry for each spawned worker (module-level for pickling)."""
    i, node_id, listen_to, peers = param
    router = Parallel()
    router._run(node_id, listen_to, peers, mq_type=mq_type, labels=labels)
    retries = 0
    while True:
        try:
            main_fn(*args, **kwargs)
            break
        except Exception:
            traceback.print_exc()
            if auto_recover and retries < max_retries:
                retries += 1
                logger.warning(f"node {node_id} crashed; auto-recover retry {retries}/{max_retries}")
                continue
            router.stop()
            raise
    router.stop()
